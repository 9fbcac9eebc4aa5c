"""Epsilon-box non-dominated archive.

Same semantics as the reference's EpsilonSort (MOEA.py:470-595, itself after
Woodruff & Herman's pareto.py): each solution maps to an epsilon-box (floor
of objectives / epsilons); a candidate is kept iff its box is not dominated
by any archived box; dominated archive members are evicted; within the same
box, the solution closer (squared distance) to the box's lower corner wins.
Implemented fresh with vectorized box comparisons.
"""

from __future__ import annotations

import math
from typing import Any, List, Optional

import numpy as np


class EpsilonSort:
    def __init__(self, epsilons):
        self.epsilons = np.asarray(
            [e if (e != 0 and not np.isnan(e)) else 1e-8 for e in epsilons], dtype=float
        )
        self.n_obj = len(self.epsilons)
        self.archive: List[np.ndarray] = []
        self.tagalongs: List[Any] = []
        self.boxes: List[np.ndarray] = []

    def _box(self, objectives: np.ndarray) -> np.ndarray:
        return np.floor(objectives / self.epsilons).astype(np.int64)

    def sortinto(self, objectives, tagalong: Optional[Any] = None) -> None:
        objectives = np.nan_to_num(np.asarray(objectives, dtype=float))
        ebox = self._box(objectives)

        if self.boxes:
            B = np.stack(self.boxes)  # (A, d)
            le = B <= ebox[None, :]
            lt = B < ebox[None, :]
            ge = B >= ebox[None, :]
            gt = B > ebox[None, :]
            a_dominates = le.all(axis=1) & lt.any(axis=1)
            s_dominates = ge.all(axis=1) & gt.any(axis=1)
            same_box = (B == ebox[None, :]).all(axis=1)

            if a_dominates.any():
                return  # candidate dominated by an archived box

            same_idx = np.where(same_box)[0]
            if len(same_idx):
                ai = int(same_idx[0])
                corner = ebox * self.epsilons
                sdist = float(((objectives - corner) ** 2).sum())
                adist = float(((self.archive[ai] - corner) ** 2).sum())
                if adist < sdist:
                    return
                self._remove(ai)
                # fall through: candidate replaces same-box occupant; other
                # members can't also be dominated (they weren't before)
                self._add(objectives, tagalong, ebox)
                return

            if s_dominates.any():
                for ai in sorted(np.where(s_dominates)[0], reverse=True):
                    self._remove(int(ai))

        self._add(objectives, tagalong, ebox)

    def _add(self, objectives, tagalong, ebox):
        self.archive.append(objectives)
        self.tagalongs.append(tagalong)
        self.boxes.append(ebox)

    def _remove(self, index: int):
        self.archive.pop(index)
        self.tagalongs.pop(index)
        self.boxes.pop(index)
