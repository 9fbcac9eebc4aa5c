"""HDF5 storage schema (placeholder — implemented by the native backend)."""


def _nyi(*a, **k):
    raise NotImplementedError("HDF5 storage backend not yet built")


init_h5 = save_to_h5 = init_from_h5 = _nyi
save_surrogate_evals_to_h5 = save_optimizer_params_to_h5 = save_stats_to_h5 = _nyi
