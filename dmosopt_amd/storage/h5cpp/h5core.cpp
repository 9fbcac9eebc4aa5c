// Native HDF5 storage core for dmosopt_amd (pybind11 + libhdf5).
//
// h5py is not part of this framework's dependency set; this module provides
// the minimal, h5py-file-compatible primitives the dmosopt HDF5 schema
// needs (reference dmosopt.py:1473-2349): committed enum / compound types
// with explicit (numpy-packed) member offsets, chunked resizable 1-D
// datasets of those types, raw-bytes append/read (memory layout == file
// layout), scalar/array/string datasets, and type introspection for
// restore. Bool fields are stored as h5py does (int8 enum {FALSE,TRUE}).

#include <hdf5.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <cstring>
#include <map>
#include <stdexcept>
#include <string>
#include <vector>

namespace py = pybind11;

static void check(herr_t status, const char* what) {
  if (status < 0) throw std::runtime_error(std::string("HDF5 error in ") + what);
}

static hid_t check_id(hid_t id, const char* what) {
  if (id < 0) throw std::runtime_error(std::string("HDF5 error in ") + what);
  return id;
}

// h5py-compatible bool: enum int8 {FALSE=0, TRUE=1}
static hid_t make_bool_type() {
  hid_t t = H5Tenum_create(H5T_NATIVE_INT8);
  int8_t v0 = 0, v1 = 1;
  H5Tenum_insert(t, "FALSE", &v0);
  H5Tenum_insert(t, "TRUE", &v1);
  return t;
}

struct H5File {
  hid_t fid = -1;
  std::string path;

  H5File(const std::string& p, const std::string& mode) : path(p) {
    H5E_auto2_t old_func;
    void* old_data;
    H5Eget_auto2(H5E_DEFAULT, &old_func, &old_data);
    H5Eset_auto2(H5E_DEFAULT, nullptr, nullptr);  // silence expected probes
    // STRONG close degree: H5Fclose force-releases any stray object handle
    // so the library's atexit teardown never spins on open ids.
    hid_t fapl = H5Pcreate(H5P_FILE_ACCESS);
    H5Pset_fclose_degree(fapl, H5F_CLOSE_STRONG);
    if (mode == "r") {
      fid = H5Fopen(p.c_str(), H5F_ACC_RDONLY, fapl);
    } else if (mode == "a" || mode == "r+") {
      fid = H5Fopen(p.c_str(), H5F_ACC_RDWR, fapl);
      if (fid < 0) fid = H5Fcreate(p.c_str(), H5F_ACC_EXCL, H5P_DEFAULT, fapl);
    } else if (mode == "w") {
      fid = H5Fcreate(p.c_str(), H5F_ACC_TRUNC, H5P_DEFAULT, fapl);
    }
    H5Pclose(fapl);
    H5Eset_auto2(H5E_DEFAULT, old_func, old_data);
    check_id(fid, ("open " + p).c_str());
  }

  void close() {
    if (fid >= 0) {
      H5Fclose(fid);
      fid = -1;
    }
  }
  ~H5File() { close(); }

  bool has(const std::string& name) {
    // supports nested paths: check each component
    std::string cur;
    size_t start = 0;
    while (start < name.size()) {
      size_t slash = name.find('/', start);
      std::string comp = name.substr(start, slash == std::string::npos ? std::string::npos : slash - start);
      if (!comp.empty()) {
        cur += "/" + comp;
        htri_t ex = H5Lexists(fid, cur.c_str(), H5P_DEFAULT);
        if (ex <= 0) return false;
      }
      if (slash == std::string::npos) break;
      start = slash + 1;
    }
    return true;
  }

  void create_group(const std::string& name) {
    if (has(name)) return;
    hid_t lcpl = H5Pcreate(H5P_LINK_CREATE);
    H5Pset_create_intermediate_group(lcpl, 1);
    hid_t g = check_id(
        H5Gcreate2(fid, name.c_str(), lcpl, H5P_DEFAULT, H5P_DEFAULT),
        "create_group");
    H5Gclose(g);
    H5Pclose(lcpl);
  }

  // ---- type construction ------------------------------------------------
  hid_t resolve_base_type(const std::string& code) {
    if (code == "f4") return H5Tcopy(H5T_NATIVE_FLOAT);
    if (code == "f8") return H5Tcopy(H5T_NATIVE_DOUBLE);
    if (code == "u2") return H5Tcopy(H5T_NATIVE_UINT16);
    if (code == "u4") return H5Tcopy(H5T_NATIVE_UINT32);
    if (code == "i4") return H5Tcopy(H5T_NATIVE_INT32);
    if (code == "i8") return H5Tcopy(H5T_NATIVE_INT64);
    if (code == "u1") return H5Tcopy(H5T_NATIVE_UINT8);
    if (code == "b1") return make_bool_type();
    if (code.rfind("S", 0) == 0) {
      size_t len = std::stoul(code.substr(1));
      hid_t t = H5Tcopy(H5T_C_S1);
      H5Tset_size(t, len);
      H5Tset_strpad(t, H5T_STR_NULLPAD);
      return t;
    }
    // committed type path
    if (has(code)) return check_id(H5Topen2(fid, code.c_str(), H5P_DEFAULT), "topen");
    throw std::runtime_error("unknown type code: " + code);
  }

  void commit_enum(const std::string& path, const std::vector<std::string>& names,
                   const std::vector<long>& values) {
    if (has(path)) return;
    hid_t t = H5Tenum_create(H5T_NATIVE_UINT16);
    for (size_t i = 0; i < names.size(); ++i) {
      uint16_t v = (uint16_t)values[i];
      H5Tenum_insert(t, names[i].c_str(), &v);
    }
    hid_t lcpl = H5Pcreate(H5P_LINK_CREATE);
    H5Pset_create_intermediate_group(lcpl, 1);
    check(H5Tcommit2(fid, path.c_str(), t, lcpl, H5P_DEFAULT, H5P_DEFAULT),
          "commit_enum");
    H5Pclose(lcpl);
    H5Tclose(t);
  }

  // members: (name, offset, type_code, n_elements) — n_elements > 1 makes a
  // 1-D array member (e.g. the path components S128 x 10)
  void commit_compound(
      const std::string& path, size_t total_size,
      const std::vector<std::tuple<std::string, size_t, std::string, size_t>>& members) {
    if (has(path)) return;
    hid_t t = H5Tcreate(H5T_COMPOUND, total_size);
    for (auto& m : members) {
      hid_t base = resolve_base_type(std::get<2>(m));
      hid_t memt = base;
      size_t nel = std::get<3>(m);
      if (nel > 1) {
        hsize_t dims[1] = {(hsize_t)nel};
        memt = H5Tarray_create2(base, 1, dims);
      }
      check(H5Tinsert(t, std::get<0>(m).c_str(), std::get<1>(m), memt), "tinsert");
      if (memt != base) H5Tclose(memt);
      H5Tclose(base);
    }
    hid_t lcpl = H5Pcreate(H5P_LINK_CREATE);
    H5Pset_create_intermediate_group(lcpl, 1);
    check(H5Tcommit2(fid, path.c_str(), t, lcpl, H5P_DEFAULT, H5P_DEFAULT),
          "commit_compound");
    H5Pclose(lcpl);
    H5Tclose(t);
  }

  // ---- datasets ----------------------------------------------------------
  void create_dataset(const std::string& path, const std::string& type_code,
                      long initial_rows, long max_rows) {
    if (has(path)) return;
    hid_t t = resolve_base_type(type_code);
    hsize_t dims[1] = {(hsize_t)initial_rows};
    hsize_t maxdims[1] = {max_rows < 0 ? H5S_UNLIMITED : (hsize_t)max_rows};
    hid_t space = H5Screate_simple(1, dims, maxdims);
    hid_t dcpl = H5Pcreate(H5P_DATASET_CREATE);
    size_t tsize = H5Tget_size(t);
    hsize_t chunk[1] = {(hsize_t)std::max<size_t>(1, 65536 / std::max<size_t>(tsize, 1))};
    H5Pset_chunk(dcpl, 1, chunk);
    hid_t lcpl = H5Pcreate(H5P_LINK_CREATE);
    H5Pset_create_intermediate_group(lcpl, 1);
    hid_t d = check_id(
        H5Dcreate2(fid, path.c_str(), t, space, lcpl, dcpl, H5P_DEFAULT),
        "dcreate");
    H5Dclose(d);
    H5Pclose(lcpl);
    H5Pclose(dcpl);
    H5Sclose(space);
    H5Tclose(t);
  }

  long dataset_rows(const std::string& path) {
    hid_t d = check_id(H5Dopen2(fid, path.c_str(), H5P_DEFAULT), "dopen");
    hid_t space = H5Dget_space(d);
    hsize_t dims[8];
    int nd = H5Sget_simple_extent_dims(space, dims, nullptr);
    H5Sclose(space);
    H5Dclose(d);
    return nd >= 1 ? (long)dims[0] : 0;
  }

  // append raw rows (memory layout == file layout by construction)
  void append_rows(const std::string& path, py::bytes data, long n_rows) {
    hid_t d = check_id(H5Dopen2(fid, path.c_str(), H5P_DEFAULT), "dopen");
    hid_t t = H5Dget_type(d);
    hid_t space = H5Dget_space(d);
    hsize_t dims[1];
    H5Sget_simple_extent_dims(space, dims, nullptr);
    H5Sclose(space);
    hsize_t newdims[1] = {dims[0] + (hsize_t)n_rows};
    check(H5Dset_extent(d, newdims), "set_extent");
    hid_t fspace = H5Dget_space(d);
    hsize_t start[1] = {dims[0]};
    hsize_t count[1] = {(hsize_t)n_rows};
    H5Sselect_hyperslab(fspace, H5S_SELECT_SET, start, nullptr, count, nullptr);
    hid_t mspace = H5Screate_simple(1, count, nullptr);
    std::string buf = data;
    if (buf.size() != (size_t)n_rows * H5Tget_size(t))
      throw std::runtime_error(
          "append_rows: buffer size " + std::to_string(buf.size()) +
          " != n_rows * type_size " +
          std::to_string((size_t)n_rows * H5Tget_size(t)) + " for " + path);
    check(H5Dwrite(d, t, mspace, fspace, H5P_DEFAULT, buf.data()), "dwrite");
    H5Sclose(mspace);
    H5Sclose(fspace);
    H5Tclose(t);
    H5Dclose(d);
  }

  void write_rows(const std::string& path, py::bytes data, long n_rows) {
    // overwrite rows [0, n_rows)
    hid_t d = check_id(H5Dopen2(fid, path.c_str(), H5P_DEFAULT), "dopen");
    hid_t t = H5Dget_type(d);
    hsize_t dims[1] = {(hsize_t)n_rows};
    check(H5Dset_extent(d, dims), "set_extent");
    hid_t fspace = H5Dget_space(d);
    hid_t mspace = H5Screate_simple(1, dims, nullptr);
    std::string buf = data;
    if (buf.size() != (size_t)n_rows * H5Tget_size(t))
      throw std::runtime_error(
          "write_rows: buffer size " + std::to_string(buf.size()) +
          " != n_rows * type_size " +
          std::to_string((size_t)n_rows * H5Tget_size(t)) + " for " + path);
    check(H5Dwrite(d, t, mspace, fspace, H5P_DEFAULT, buf.data()), "dwrite");
    H5Sclose(mspace);
    H5Sclose(fspace);
    H5Tclose(t);
    H5Dclose(d);
  }

  py::bytes read_rows(const std::string& path) {
    hid_t d = check_id(H5Dopen2(fid, path.c_str(), H5P_DEFAULT), "dopen");
    hid_t t = H5Dget_type(d);
    hid_t space = H5Dget_space(d);
    hssize_t n = H5Sget_simple_extent_npoints(space);
    size_t tsize = H5Tget_size(t);
    std::string buf(((size_t)n) * tsize, '\0');
    if (n > 0)
      check(H5Dread(d, t, H5S_ALL, H5S_ALL, H5P_DEFAULT, buf.data()), "dread");
    H5Sclose(space);
    H5Tclose(t);
    H5Dclose(d);
    return py::bytes(buf);
  }

  // ---- scalars / simple arrays ------------------------------------------
  void write_string(const std::string& path, const std::string& value) {
    if (has(path)) return;
    hid_t t = H5Tcopy(H5T_C_S1);
    H5Tset_size(t, H5T_VARIABLE);
    H5Tset_cset(t, H5T_CSET_UTF8);
    hid_t space = H5Screate(H5S_SCALAR);
    hid_t lcpl = H5Pcreate(H5P_LINK_CREATE);
    H5Pset_create_intermediate_group(lcpl, 1);
    hid_t d = check_id(
        H5Dcreate2(fid, path.c_str(), t, space, lcpl, H5P_DEFAULT, H5P_DEFAULT),
        "dcreate str");
    H5Pclose(lcpl);
    const char* cstr = value.c_str();
    check(H5Dwrite(d, t, H5S_ALL, H5S_ALL, H5P_DEFAULT, &cstr), "dwrite str");
    H5Dclose(d);
    H5Sclose(space);
    H5Tclose(t);
  }

  void write_simple(const std::string& path, const std::string& type_code,
                    py::bytes data, const std::vector<long>& shape) {
    if (has(path)) return;
    hid_t t = resolve_base_type(type_code);
    std::vector<hsize_t> dims(shape.begin(), shape.end());
    hid_t space = shape.empty()
                      ? H5Screate(H5S_SCALAR)
                      : H5Screate_simple((int)dims.size(), dims.data(), nullptr);
    hid_t lcpl = H5Pcreate(H5P_LINK_CREATE);
    H5Pset_create_intermediate_group(lcpl, 1);
    hid_t d = check_id(
        H5Dcreate2(fid, path.c_str(), t, space, lcpl, H5P_DEFAULT, H5P_DEFAULT),
        "dcreate simple");
    H5Pclose(lcpl);
    std::string buf = data;
    check(H5Dwrite(d, t, H5S_ALL, H5S_ALL, H5P_DEFAULT, buf.data()), "dwrite simple");
    H5Dclose(d);
    H5Sclose(space);
    H5Tclose(t);
  }

  // ---- introspection ------------------------------------------------------
  // returns a python description of a datatype: for compound -> list of
  // (name, offset, desc, nel); enum -> ("enum", base_code, names, values);
  // string -> ("S", size); numeric -> code
  py::object describe_type(hid_t t) {
    H5T_class_t cls = H5Tget_class(t);
    if (cls == H5T_COMPOUND) {
      py::list members;
      int n = H5Tget_nmembers(t);
      for (int i = 0; i < n; ++i) {
        char* nm = H5Tget_member_name(t, (unsigned)i);
        size_t off = H5Tget_member_offset(t, (unsigned)i);
        hid_t mt = H5Tget_member_type(t, (unsigned)i);
        size_t nel = 1;
        hid_t base = mt;
        bool is_array = (H5Tget_class(mt) == H5T_ARRAY);
        if (is_array) {
          hsize_t adims[4];
          int nd = H5Tget_array_ndims(mt);
          H5Tget_array_dims2(mt, adims);
          nel = 1;
          for (int k = 0; k < nd; ++k) nel *= (size_t)adims[k];
          base = H5Tget_super(mt);
        }
        members.append(py::make_tuple(std::string(nm), off, describe_type(base),
                                      nel));
        if (is_array) H5Tclose(base);
        H5Tclose(mt);
        H5free_memory(nm);
      }
      return py::make_tuple(std::string("compound"), H5Tget_size(t), members);
    }
    if (cls == H5T_ENUM) {
      py::list names;
      py::list values;
      int n = H5Tget_nmembers(t);
      hid_t super_t = H5Tget_super(t);
      size_t ssize = H5Tget_size(super_t);
      for (int i = 0; i < n; ++i) {
        char* nm = H5Tget_member_name(t, (unsigned)i);
        long long val = 0;
        H5Tget_member_value(t, (unsigned)i, &val);  // little-endian ok
        names.append(std::string(nm));
        values.append((long)val);
        H5free_memory(nm);
      }
      H5Tclose(super_t);
      return py::make_tuple(std::string("enum"), (long)ssize, names, values);
    }
    if (cls == H5T_STRING) {
      return py::make_tuple(std::string("string"), (long)H5Tget_size(t));
    }
    if (cls == H5T_FLOAT) {
      return py::str(H5Tget_size(t) == 4 ? "f4" : "f8");
    }
    if (cls == H5T_INTEGER) {
      size_t s = H5Tget_size(t);
      bool sgn = H5Tget_sign(t) == H5T_SGN_2;
      std::string code = (sgn ? "i" : "u") + std::to_string(s);
      return py::str(code);
    }
    return py::str("unknown");
  }

  py::object dataset_type(const std::string& path) {
    hid_t d = check_id(H5Dopen2(fid, path.c_str(), H5P_DEFAULT), "dopen");
    hid_t t = H5Dget_type(d);
    py::object desc = describe_type(t);
    H5Tclose(t);
    H5Dclose(d);
    return desc;
  }

  py::object committed_type(const std::string& path) {
    hid_t t = check_id(H5Topen2(fid, path.c_str(), H5P_DEFAULT), "topen");
    py::object desc = describe_type(t);
    H5Tclose(t);
    return desc;
  }

  std::string read_string(const std::string& path) {
    hid_t d = check_id(H5Dopen2(fid, path.c_str(), H5P_DEFAULT), "dopen");
    hid_t t = H5Dget_type(d);
    std::string out;
    if (H5Tis_variable_str(t) > 0) {
      char* ptr = nullptr;
      hid_t mt = H5Tcopy(t);  // same cset/class as the file type
      check(H5Dread(d, mt, H5S_ALL, H5S_ALL, H5P_DEFAULT, &ptr), "dread str");
      if (ptr) {
        out = ptr;
        H5free_memory(ptr);
      }
      H5Tclose(mt);
    } else {
      size_t sz = H5Tget_size(t);
      out.resize(sz);
      check(H5Dread(d, t, H5S_ALL, H5S_ALL, H5P_DEFAULT, out.data()), "dread fstr");
    }
    H5Tclose(t);
    H5Dclose(d);
    return out;
  }

  std::vector<std::string> list_group(const std::string& path) {
    std::vector<std::string> out;
    hid_t g = check_id(H5Gopen2(fid, path.c_str(), H5P_DEFAULT), "gopen");
    H5G_info_t info;
    H5Gget_info(g, &info);
    for (hsize_t i = 0; i < info.nlinks; ++i) {
      ssize_t len = H5Lget_name_by_idx(g, ".", H5_INDEX_NAME, H5_ITER_NATIVE, i,
                                       nullptr, 0, H5P_DEFAULT);
      std::string name(len, '\0');
      H5Lget_name_by_idx(g, ".", H5_INDEX_NAME, H5_ITER_NATIVE, i, name.data(),
                         len + 1, H5P_DEFAULT);
      out.push_back(name);
    }
    H5Gclose(g);
    return out;
  }
};

PYBIND11_MODULE(_h5core, m) {
  m.doc() = "Native HDF5 storage core (libhdf5) for dmosopt_amd";
  // Errors surface as Python exceptions via check()/check_id(); the default
  // stderr error stack (incl. teardown diagnostics) is noise here.
  // Every file is closed explicitly with H5F_CLOSE_STRONG, so HDF5's own
  // atexit teardown has nothing to do — and on a GPU box it runs in
  // undefined order against the HIP runtime's teardown (observed exit-time
  // SIGSEGV after a run that both used the GPU and saved results). MUST be
  // the first HDF5 call: any other call initializes the library and
  // registers the atexit handler.
  H5dont_atexit();
  H5Eset_auto2(H5E_DEFAULT, nullptr, nullptr);
  py::class_<H5File>(m, "H5File")
      .def(py::init<const std::string&, const std::string&>())
      .def("close", &H5File::close)
      .def("has", &H5File::has)
      .def("create_group", &H5File::create_group)
      .def("commit_enum", &H5File::commit_enum)
      .def("commit_compound", &H5File::commit_compound)
      .def("create_dataset", &H5File::create_dataset, py::arg("path"),
           py::arg("type_code"), py::arg("initial_rows") = 0,
           py::arg("max_rows") = -1)
      .def("dataset_rows", &H5File::dataset_rows)
      .def("append_rows", &H5File::append_rows)
      .def("write_rows", &H5File::write_rows)
      .def("read_rows", &H5File::read_rows)
      .def("write_string", &H5File::write_string)
      .def("write_simple", &H5File::write_simple)
      .def("dataset_type", &H5File::dataset_type)
      .def("committed_type", &H5File::committed_type)
      .def("read_string", &H5File::read_string)
      .def("list_group", &H5File::list_group);
}
