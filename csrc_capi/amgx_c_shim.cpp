// libamgx_amd.so — the linkable AMGX_* C ABI (reference include/amgx_c.h,
// src/amgx_c.cu 5,358 LoC).
//
// First-cut architecture (VERDICT r01 "Ship a real C ABI"): this shared
// library embeds a CPython interpreter and routes every entry point into
// amgx_amd.capi — the Python control plane that already mirrors the full
// AMGX_* surface with handle/RC discipline — which in turn drives the
// hand-written gfx950 HIP kernel extension (amgx_amd/csrc).  C and Fortran
// hosts #include <amgx_c.h>, link -lamgx_amd, and get the reference calling
// convention; the numeric hot path stays native HIP end to end.
//
// Threading: the embedded interpreter's GIL is released after init; every
// entry point re-acquires it (PyGILState), so the ABI is callable from any
// host thread (reference _lock/_unlock per call, src/amgx_c.cu).

#include <Python.h>

#include <dlfcn.h>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <string>

#include "../include/amgx_c.h"
#include "../include/amgx_eig_c.h"

namespace {

PyObject *g_capi = nullptr;     // amgx_amd.capi module
PyObject *g_np = nullptr;       // numpy module
PyThreadState *g_main_tstate = nullptr;
bool g_we_initialized = false;
AMGX_print_callback g_print_cb = nullptr;

struct Gil {
    PyGILState_STATE st;
    Gil() { st = PyGILState_Ensure(); }
    ~Gil() { PyGILState_Release(st); }
};

// amgx_amd.capi RC numbering -> reference AMGX_RC numbering (they diverge
// from code 6 up: the reference interleaves THRUST_FAILURE/NO_MEMORY).
AMGX_RC map_rc(long rc) {
    static const AMGX_RC table[] = {
        AMGX_RC_OK, AMGX_RC_BAD_PARAMETERS, AMGX_RC_UNKNOWN,
        AMGX_RC_NOT_SUPPORTED_TARGET, AMGX_RC_NOT_SUPPORTED_BLOCKSIZE,
        AMGX_RC_CUDA_FAILURE, AMGX_RC_IO_ERROR, AMGX_RC_BAD_MODE,
        AMGX_RC_CORE, AMGX_RC_PLUGIN, AMGX_RC_BAD_CONFIGURATION,
        AMGX_RC_NOT_IMPLEMENTED, AMGX_RC_LICENSE_NOT_FOUND,
        AMGX_RC_INTERNAL};
    if (rc >= 0 && rc < (long)(sizeof(table) / sizeof(table[0])))
        return table[rc];
    return AMGX_RC_UNKNOWN;
}

const char *mode_str(AMGX_Mode mode) {
    switch (mode) {
        case AMGX_mode_hDDI: return "hDDI";
        case AMGX_mode_hDFI: return "hDFI";
        case AMGX_mode_hFFI: return "hFFI";
        case AMGX_mode_dDDI: return "dDDI";
        case AMGX_mode_dDFI: return "dDFI";
        case AMGX_mode_dFFI: return "dFFI";
        default: return nullptr;
    }
}

// element width of the MATRIX scalar type for a mode
size_t mat_elem_size(AMGX_Mode mode) {
    switch (mode) {
        case AMGX_mode_hDDI: case AMGX_mode_dDDI: return 8;
        default: return 4;   // *F* matrix modes
    }
}
const char *mat_dtype(AMGX_Mode mode) {
    return mat_elem_size(mode) == 8 ? "float64" : "float32";
}
// vector precision
size_t vec_elem_size(AMGX_Mode mode) {
    switch (mode) {
        case AMGX_mode_hFFI: case AMGX_mode_dFFI: return 4;
        default: return 8;   // *D* vector modes
    }
}
const char *vec_dtype(AMGX_Mode mode) {
    return vec_elem_size(mode) == 8 ? "float64" : "float32";
}

void report_py_error(const char *where) {
    if (!PyErr_Occurred()) return;
    PyObject *type, *value, *tb;
    PyErr_Fetch(&type, &value, &tb);
    PyObject *s = value ? PyObject_Str(value) : nullptr;
    const char *msg = s ? PyUnicode_AsUTF8(s) : "unknown";
    char buf[1024];
    snprintf(buf, sizeof(buf), "AMGX C-ABI error in %s: %s\n", where,
             msg ? msg : "unknown");
    if (g_print_cb) g_print_cb(buf, (int)strlen(buf));
    else fputs(buf, stderr);
    Py_XDECREF(s);
    Py_XDECREF(type); Py_XDECREF(value); Py_XDECREF(tb);
    PyErr_Clear();
}

// Call amgx_amd.capi.<name>(*args).  args is a NEW reference (stolen).
// Returns new reference or nullptr.
PyObject *call_capi(const char *name, PyObject *args) {
    if (!g_capi) { Py_XDECREF(args); return nullptr; }
    PyObject *fn = PyObject_GetAttrString(g_capi, name);
    if (!fn) { report_py_error(name); Py_XDECREF(args); return nullptr; }
    PyObject *res = PyObject_CallObject(fn, args);
    Py_DECREF(fn);
    Py_XDECREF(args);
    if (!res) report_py_error(name);
    return res;
}

// Result may be an int RC or a tuple whose [0] is the RC.  Consumes res.
// extras: borrowed pointers into the tuple written to out[i] as NEW refs.
AMGX_RC unpack_rc(PyObject *res, PyObject **out = nullptr, int n_out = 0,
                  const char *where = "") {
    if (!res) return AMGX_RC_INTERNAL;
    long rc;
    if (PyTuple_Check(res)) {
        rc = PyLong_AsLong(PyTuple_GetItem(res, 0));
        for (int i = 0; i < n_out; ++i) {
            PyObject *item = (Py_ssize_t)(i + 1) < PyTuple_Size(res)
                                 ? PyTuple_GetItem(res, i + 1)
                                 : Py_None;
            Py_INCREF(item);
            out[i] = item;
        }
    } else {
        rc = PyLong_AsLong(res);
        for (int i = 0; i < n_out; ++i) { Py_INCREF(Py_None); out[i] = Py_None; }
    }
    Py_DECREF(res);
    if (PyErr_Occurred()) { report_py_error(where); return AMGX_RC_INTERNAL; }
    return map_rc(rc);
}

// numpy array COPY of raw C memory (count elements of dtype)
PyObject *np_from_mem(const void *ptr, Py_ssize_t count, const char *dtype,
                      size_t elem) {
    if (!ptr) Py_RETURN_NONE;
    PyObject *mv = PyMemoryView_FromMemory((char *)ptr, count * elem,
                                           PyBUF_READ);
    if (!mv) return nullptr;
    PyObject *arr = PyObject_CallMethod(g_np, "frombuffer", "Os", mv, dtype);
    Py_DECREF(mv);
    if (!arr) return nullptr;
    PyObject *copy = PyObject_CallMethod(arr, "copy", nullptr);
    Py_DECREF(arr);
    return copy;
}

// copy a (numpy-like, buffer-protocol) object's contiguous bytes to dst
bool copy_out(PyObject *arr, void *dst) {
    if (!arr || arr == Py_None || !dst) return false;
    Py_buffer view;
    if (PyObject_GetBuffer(arr, &view, PyBUF_CONTIG_RO) != 0) {
        PyObject *c = PyObject_CallMethod(g_np, "ascontiguousarray", "O", arr);
        if (!c) { PyErr_Clear(); return false; }
        bool ok = PyObject_GetBuffer(c, &view, PyBUF_CONTIG_RO) == 0;
        Py_DECREF(c);
        if (!ok) { PyErr_Clear(); return false; }
    }
    memcpy(dst, view.buf, view.len);
    PyBuffer_Release(&view);
    return true;
}

// handle <-> PyObject: the opaque C handle IS a new reference
PyObject *obj(const void *h) { return (PyObject *)h; }

// NOTE: Py_BuildValue needs the GIL, so the helpers take a format +
// varargs and build the tuple AFTER acquiring it (building at the call
// site would run GIL-less and crash).
AMGX_RC create_genericv(const char *fn_name, void **out, const char *where,
                        const char *fmt, ...) {
    Gil gil;
    PyObject *args = nullptr;
    if (fmt) {
        va_list va;
        va_start(va, fmt);
        args = Py_VaBuildValue(fmt, va);
        va_end(va);
        if (!args) { report_py_error(where); *out = nullptr;
                     return AMGX_RC_INTERNAL; }
    }
    PyObject *res = call_capi(fn_name, args);
    PyObject *handle = nullptr;
    AMGX_RC rc = unpack_rc(res, &handle, 1, where);
    if (rc == AMGX_RC_OK && handle && handle != Py_None) {
        *out = (void *)handle;    // keep the new reference in the C handle
    } else {
        Py_XDECREF(handle);
        if (rc == AMGX_RC_OK) rc = AMGX_RC_INTERNAL;
        *out = nullptr;
    }
    return rc;
}

AMGX_RC simple_callv(const char *fn_name, const char *where, const char *fmt,
                     ...) {
    Gil gil;
    PyObject *args = nullptr;
    if (fmt) {
        va_list va;
        va_start(va, fmt);
        args = Py_VaBuildValue(fmt, va);
        va_end(va);
        if (!args) { report_py_error(where); return AMGX_RC_INTERNAL; }
    }
    return unpack_rc(call_capi(fn_name, args), nullptr, 0, where);
}

// Python print-callback trampoline -> registered C function pointer
PyObject *py_print_trampoline(PyObject *, PyObject *arg) {
    if (g_print_cb) {
        Py_ssize_t len = 0;
        const char *msg = PyUnicode_AsUTF8AndSize(arg, &len);
        if (msg) g_print_cb(msg, (int)len);
    }
    Py_RETURN_NONE;
}
PyMethodDef g_print_def = {"amgx_c_print", py_print_trampoline, METH_O,
                           nullptr};

std::string find_pyroot() {
    if (const char *env = getenv("AMGX_AMD_PYROOT")) return env;
    Dl_info info;
    if (dladdr((void *)&AMGX_initialize, &info) && info.dli_fname) {
        std::string p(info.dli_fname);
        auto slash = p.rfind('/');
        if (slash != std::string::npos) {
            p = p.substr(0, slash);          // .../csrc_capi or .../lib
            auto slash2 = p.rfind('/');
            if (slash2 != std::string::npos) return p.substr(0, slash2);
        }
    }
    return ".";
}

}  // namespace

extern "C" {

AMGX_RC AMGX_initialize(void) {
    if (!Py_IsInitialized()) {
        Py_InitializeEx(0);
        g_we_initialized = true;
    }
    PyGILState_STATE st = PyGILState_Ensure();
    // make the amgx_amd package importable next to this library
    std::string root = find_pyroot();
    PyObject *sys_path = PySys_GetObject("path");
    if (sys_path) {
        PyObject *p = PyUnicode_FromString(root.c_str());
        if (p) { PyList_Insert(sys_path, 0, p); Py_DECREF(p); }
    }
    if (!g_np) g_np = PyImport_ImportModule("numpy");
    if (!g_capi) g_capi = PyImport_ImportModule("amgx_amd.capi");
    bool ok = g_capi && g_np;
    if (!ok) report_py_error("AMGX_initialize(import)");
    AMGX_RC rc = AMGX_RC_INTERNAL;
    if (ok)
        rc = unpack_rc(call_capi("AMGX_initialize", nullptr), nullptr, 0,
                       "AMGX_initialize");
    PyGILState_Release(st);
    // release the GIL for the host app; entry points re-acquire per call
    if (g_we_initialized && !g_main_tstate)
        g_main_tstate = PyEval_SaveThread();
    return ok ? rc : AMGX_RC_INTERNAL;
}

AMGX_RC AMGX_initialize_plugins(void) {
    return simple_callv("AMGX_initialize_plugins", "init_plugins", nullptr);
}

AMGX_RC AMGX_finalize(void) {
    Gil gil;
    AMGX_RC rc = unpack_rc(call_capi("AMGX_finalize", nullptr), nullptr, 0,
                           "AMGX_finalize");
    // the interpreter stays up (cheap, and torch teardown at exit is safer
    // left to the process); matches reference allowing re-initialization
    return rc;
}

AMGX_RC AMGX_finalize_plugins(void) {
    return simple_callv("AMGX_finalize_plugins", "fin_plugins", nullptr);
}

void AMGX_abort(AMGX_resources_handle, int err) {
    fprintf(stderr, "AMGX_abort(%d)\n", err);
    exit(err);
}

AMGX_RC AMGX_pin_memory(void *, unsigned int) { return AMGX_RC_OK; }
AMGX_RC AMGX_unpin_memory(void *) { return AMGX_RC_OK; }

AMGX_RC AMGX_install_signal_handler(void) {
    return simple_callv("AMGX_install_signal_handler", "sig", nullptr);
}
AMGX_RC AMGX_reset_signal_handler(void) {
    return simple_callv("AMGX_reset_signal_handler", "sig", nullptr);
}

AMGX_RC AMGX_register_print_callback(AMGX_print_callback func) {
    Gil gil;
    g_print_cb = func;
    PyObject *cb = PyCFunction_New(&g_print_def, nullptr);
    if (!cb) return AMGX_RC_INTERNAL;
    return unpack_rc(call_capi("AMGX_register_print_callback",
                               Py_BuildValue("(N)", cb)),
                     nullptr, 0, "register_print_callback");
}

AMGX_RC AMGX_get_api_version(int *major, int *minor) {
    Gil gil;
    PyObject *out[2] = {nullptr, nullptr};
    AMGX_RC rc = unpack_rc(call_capi("AMGX_get_api_version", nullptr), out, 2,
                           "get_api_version");
    if (rc == AMGX_RC_OK) {
        if (major) *major = (int)PyLong_AsLong(out[0]);
        if (minor) *minor = (int)PyLong_AsLong(out[1]);
    }
    Py_XDECREF(out[0]); Py_XDECREF(out[1]);
    return rc;
}

AMGX_RC AMGX_get_build_info_strings(char **version, char **date,
                                    char **time) {
    static char v[128], d[128], t[128];
    Gil gil;
    PyObject *out[3] = {nullptr, nullptr, nullptr};
    AMGX_RC rc = unpack_rc(call_capi("AMGX_get_build_info_strings", nullptr),
                           out, 3, "build_info");
    if (rc == AMGX_RC_OK) {
        snprintf(v, sizeof(v), "%s", PyUnicode_AsUTF8(out[0]));
        snprintf(d, sizeof(d), "%s", PyUnicode_AsUTF8(out[1]));
        snprintf(t, sizeof(t), "%s", PyUnicode_AsUTF8(out[2]));
        if (version) *version = v;
        if (date) *date = d;
        if (time) *time = t;
    }
    for (auto *o : out) Py_XDECREF(o);
    return rc;
}

AMGX_RC AMGX_get_error_string(AMGX_RC err, char *buf, int buf_len) {
    static const char *strings[] = {
        "success", "bad parameters", "unknown error",
        "unsupported target", "unsupported block size", "HIP failure",
        "rocPRIM failure", "out of memory", "I/O error", "bad mode",
        "core error", "plugin error", "bad configuration",
        "not implemented", "license not found", "internal error"};
    const char *s = (err >= 0 && err <= 15) ? strings[err] : "invalid code";
    if (buf && buf_len > 0) snprintf(buf, (size_t)buf_len, "%s", s);
    return AMGX_RC_OK;
}

/* ------------------------------------------------------------- config */
AMGX_RC AMGX_config_create(AMGX_config_handle *cfg, const char *options) {
    return create_genericv("AMGX_config_create", (void **)cfg, "config_create",
                          "(s)", options);
}

AMGX_RC AMGX_config_create_from_file(AMGX_config_handle *cfg,
                                     const char *param_file) {
    return create_genericv("AMGX_config_create_from_file", (void **)cfg, "config_create_from_file",
                          "(s)", param_file);
}

AMGX_RC AMGX_config_create_from_file_and_string(AMGX_config_handle *cfg,
                                                const char *param_file,
                                                const char *options) {
    return create_genericv("AMGX_config_create_from_file_and_string", (void **)cfg, "config_create_from_file_and_string",
                          "(ss)", param_file, options);
}

AMGX_RC AMGX_config_add_parameters(AMGX_config_handle *cfg,
                                   const char *options) {
    if (!cfg || !*cfg) return AMGX_RC_BAD_PARAMETERS;
    return simple_callv("AMGX_config_add_parameters", "config_add_parameters", "(Os)",
                       obj(*cfg), options);
}

AMGX_RC AMGX_config_get_default_number_of_rings(AMGX_config_handle cfg,
                                                int *num_import_rings) {
    Gil gil;
    PyObject *out = nullptr;
    AMGX_RC rc = unpack_rc(call_capi("AMGX_config_get_default_number_of_rings",
                                     Py_BuildValue("(O)", obj(cfg))),
                           &out, 1, "default_rings");
    if (rc == AMGX_RC_OK && num_import_rings)
        *num_import_rings = (int)PyLong_AsLong(out);
    Py_XDECREF(out);
    return rc;
}

AMGX_RC AMGX_config_destroy(AMGX_config_handle cfg) {
    if (!cfg) return AMGX_RC_BAD_PARAMETERS;
    Gil gil;
    AMGX_RC rc = unpack_rc(call_capi("AMGX_config_destroy",
                                     Py_BuildValue("(O)", obj(cfg))),
                           nullptr, 0, "config_destroy");
    Py_DECREF(obj(cfg));
    return rc;
}

/* ---------------------------------------------------------- resources */
AMGX_RC AMGX_resources_create(AMGX_resources_handle *rsc,
                              AMGX_config_handle cfg, void *comm,
                              int device_num, const int *devices) {
    (void)devices;
    // comm != NULL flags a distributed context (process bootstrap is
    // torch.distributed, not MPI — SURVEY §5.8 stance)
    return create_genericv("AMGX_resources_create", (void **)rsc,
                           "resources_create", "(OOi)", obj(cfg),
                           comm ? Py_True : Py_None, device_num);
}

AMGX_RC AMGX_resources_create_simple(AMGX_resources_handle *rsc,
                                     AMGX_config_handle cfg) {
    return create_genericv("AMGX_resources_create_simple", (void **)rsc, "resources_create_simple",
                          "(O)", obj(cfg));
}

AMGX_RC AMGX_resources_destroy(AMGX_resources_handle rsc) {
    if (!rsc) return AMGX_RC_BAD_PARAMETERS;
    Gil gil;
    AMGX_RC rc = unpack_rc(call_capi("AMGX_resources_destroy",
                                     Py_BuildValue("(O)", obj(rsc))),
                           nullptr, 0, "resources_destroy");
    Py_DECREF(obj(rsc));
    return rc;
}

/* -------------------------------------------------------- distribution */
AMGX_RC AMGX_distribution_create(AMGX_distribution_handle *dist,
                                 AMGX_config_handle cfg) {
    if (cfg)
        return create_genericv("AMGX_distribution_create", (void **)dist,
                               "distribution_create", "(O)", obj(cfg));
    return create_genericv("AMGX_distribution_create", (void **)dist,
                           "distribution_create", nullptr);
}

AMGX_RC AMGX_distribution_destroy(AMGX_distribution_handle dist) {
    if (!dist) return AMGX_RC_BAD_PARAMETERS;
    Gil gil;
    AMGX_RC rc = unpack_rc(call_capi("AMGX_distribution_destroy",
                                     Py_BuildValue("(O)", obj(dist))),
                           nullptr, 0, "distribution_destroy");
    Py_DECREF(obj(dist));
    return rc;
}

AMGX_RC AMGX_distribution_set_partition_data(AMGX_distribution_handle dist,
                                             AMGX_DIST_PARTITION_INFO info,
                                             const void *partition_data) {
    Gil gil;
    // partition offsets are rank+1 64-bit entries; the Python side stores
    // whatever array we hand it.  Sizes are unknown here for the VECTOR
    // form — the Python upload path only indexes it, so ship offsets-sized
    // data for OFFSETS and let VECTOR users go through the Python API.
    if (info != AMGX_DIST_PARTITION_OFFSETS)
        return AMGX_RC_NOT_IMPLEMENTED;   // C-side size is unknowable
    // offsets length = world+1; world from torch.distributed via capi
    PyObject *res = call_capi("_c_abi_world_size", nullptr);
    long world = 1;
    if (res) { world = PyLong_AsLong(res); Py_DECREF(res); }
    else PyErr_Clear();
    PyObject *arr = np_from_mem(partition_data, world + 1, "int64", 8);
    if (!arr) return AMGX_RC_INTERNAL;
    return unpack_rc(call_capi("AMGX_distribution_set_partition_data",
                               Py_BuildValue("(OiN)", obj(dist), (int)info,
                                             arr)),
                     nullptr, 0, "set_partition_data");
}

AMGX_RC AMGX_distribution_set_32bit_colindices(AMGX_distribution_handle dist,
                                               int use32bit) {
    return simple_callv("AMGX_distribution_set_32bit_colindices", "set_32bit_colindices", "(Oi)",
                       obj(dist), use32bit);
}

/* -------------------------------------------------------------- matrix */
AMGX_RC AMGX_matrix_create(AMGX_matrix_handle *mtx, AMGX_resources_handle rsc,
                           AMGX_Mode mode) {
    const char *ms = mode_str(mode);
    if (!ms) return AMGX_RC_BAD_MODE;
    return create_genericv("AMGX_matrix_create", (void **)mtx, "matrix_create",
                          "(Os)", obj(rsc), ms);
}

AMGX_RC AMGX_matrix_destroy(AMGX_matrix_handle mtx) {
    if (!mtx) return AMGX_RC_BAD_PARAMETERS;
    Gil gil;
    AMGX_RC rc = unpack_rc(call_capi("AMGX_matrix_destroy",
                                     Py_BuildValue("(O)", obj(mtx))),
                           nullptr, 0, "matrix_destroy");
    Py_DECREF(obj(mtx));
    return rc;
}

static AMGX_Mode handle_mode(PyObject *h) {
    PyObject *m = PyObject_GetAttrString(h, "mode");
    if (!m) { PyErr_Clear(); return AMGX_mode_hDDI; }
    const char *s = PyUnicode_AsUTF8(m);
    AMGX_Mode mode = AMGX_mode_hDDI;
    if (s) {
        if (!strcmp(s, "hDDI")) mode = AMGX_mode_hDDI;
        else if (!strcmp(s, "hDFI")) mode = AMGX_mode_hDFI;
        else if (!strcmp(s, "hFFI")) mode = AMGX_mode_hFFI;
        else if (!strcmp(s, "dDDI")) mode = AMGX_mode_dDDI;
        else if (!strcmp(s, "dDFI")) mode = AMGX_mode_dDFI;
        else if (!strcmp(s, "dFFI")) mode = AMGX_mode_dFFI;
    }
    Py_DECREF(m);
    return mode;
}

AMGX_RC AMGX_matrix_upload_all(AMGX_matrix_handle mtx, int n, int nnz,
                               int block_dimx, int block_dimy,
                               const int *row_ptrs, const int *col_indices,
                               const void *data, const void *diag_data) {
    Gil gil;
    AMGX_Mode mode = handle_mode(obj(mtx));
    size_t es = mat_elem_size(mode);
    const char *dt = mat_dtype(mode);
    long bb = (long)block_dimx * block_dimy;
    PyObject *ro = np_from_mem(row_ptrs, n + 1, "int32", 4);
    PyObject *ci = np_from_mem(col_indices, nnz, "int32", 4);
    PyObject *va = np_from_mem(data, (Py_ssize_t)nnz * bb, dt, es);
    PyObject *dg = diag_data
                       ? np_from_mem(diag_data, (Py_ssize_t)n * bb, dt, es)
                       : (Py_INCREF(Py_None), Py_None);
    if (!ro || !ci || !va || !dg) {
        Py_XDECREF(ro); Py_XDECREF(ci); Py_XDECREF(va); Py_XDECREF(dg);
        return AMGX_RC_INTERNAL;
    }
    return unpack_rc(call_capi("AMGX_matrix_upload_all",
                               Py_BuildValue("(OiiiiNNNN)", obj(mtx), n, nnz,
                                             block_dimx, block_dimy, ro, ci,
                                             va, dg)),
                     nullptr, 0, "matrix_upload_all");
}

AMGX_RC AMGX_matrix_replace_coefficients(AMGX_matrix_handle mtx, int n,
                                         int nnz, const void *data,
                                         const void *diag_data) {
    Gil gil;
    AMGX_Mode mode = handle_mode(obj(mtx));
    size_t es = mat_elem_size(mode);
    const char *dt = mat_dtype(mode);
    PyObject *bdo = PyObject_GetAttrString(obj(mtx), "A");
    long bb = 1;
    if (bdo && bdo != Py_None) {
        PyObject *bd = PyObject_GetAttrString(bdo, "block_dim");
        if (bd) { bb = PyLong_AsLong(bd); bb *= bb; Py_DECREF(bd); }
    }
    Py_XDECREF(bdo);
    PyErr_Clear();
    PyObject *va = np_from_mem(data, (Py_ssize_t)nnz * bb, dt, es);
    PyObject *dg = diag_data
                       ? np_from_mem(diag_data, (Py_ssize_t)n * bb, dt, es)
                       : (Py_INCREF(Py_None), Py_None);
    if (!va || !dg) { Py_XDECREF(va); Py_XDECREF(dg); return AMGX_RC_INTERNAL; }
    return unpack_rc(call_capi("AMGX_matrix_replace_coefficients",
                               Py_BuildValue("(OiiNN)", obj(mtx), n, nnz, va,
                                             dg)),
                     nullptr, 0, "matrix_replace_coefficients");
}

AMGX_RC AMGX_matrix_get_size(const AMGX_matrix_handle mtx, int *n,
                             int *block_dimx, int *block_dimy) {
    Gil gil;
    PyObject *out[3] = {nullptr, nullptr, nullptr};
    AMGX_RC rc = unpack_rc(call_capi("AMGX_matrix_get_size",
                                     Py_BuildValue("(O)", obj(mtx))),
                           out, 3, "matrix_get_size");
    if (rc == AMGX_RC_OK) {
        if (n) *n = (int)PyLong_AsLong(out[0]);
        if (block_dimx) *block_dimx = (int)PyLong_AsLong(out[1]);
        if (block_dimy) *block_dimy = (int)PyLong_AsLong(out[2]);
    }
    for (auto *o : out) Py_XDECREF(o);
    return rc;
}

AMGX_RC AMGX_matrix_get_nnz(const AMGX_matrix_handle mtx, int *nnz) {
    Gil gil;
    PyObject *out = nullptr;
    AMGX_RC rc = unpack_rc(call_capi("AMGX_matrix_get_nnz",
                                     Py_BuildValue("(O)", obj(mtx))),
                           &out, 1, "matrix_get_nnz");
    if (rc == AMGX_RC_OK && nnz) *nnz = (int)PyLong_AsLong(out);
    Py_XDECREF(out);
    return rc;
}

AMGX_RC AMGX_matrix_download_all(const AMGX_matrix_handle mtx, int *row_ptrs,
                                 int *col_indices, void *data,
                                 void **diag_data) {
    Gil gil;
    PyObject *out[4] = {nullptr, nullptr, nullptr, nullptr};
    AMGX_RC rc = unpack_rc(call_capi("AMGX_matrix_download_all",
                                     Py_BuildValue("(O)", obj(mtx))),
                           out, 4, "matrix_download_all");
    if (rc == AMGX_RC_OK) {
        // row_offsets may come back int32 or int64; normalize to int32
        PyObject *ro32 = PyObject_CallMethod(out[0], "astype", "s", "int32");
        PyObject *ci32 = PyObject_CallMethod(out[1], "astype", "s", "int32");
        if (!copy_out(ro32 ? ro32 : out[0], row_ptrs) ||
            !copy_out(ci32 ? ci32 : out[1], col_indices) ||
            !copy_out(out[2], data))
            rc = AMGX_RC_INTERNAL;
        Py_XDECREF(ro32); Py_XDECREF(ci32);
        // external diagonal is always folded into CSR at upload here
        if (diag_data) *diag_data = nullptr;
    }
    for (auto *o : out) Py_XDECREF(o);
    return rc;
}

AMGX_RC AMGX_matrix_vector_multiply(AMGX_matrix_handle mtx,
                                    AMGX_vector_handle x,
                                    AMGX_vector_handle y) {
    return simple_callv("AMGX_matrix_vector_multiply", "matrix_vector_multiply", "(OOO)",
                       obj(mtx), obj(x), obj(y));
}

AMGX_RC AMGX_matrix_set_boundary_separation(AMGX_matrix_handle mtx,
                                            int boundary_separation) {
    return simple_callv("AMGX_matrix_set_boundary_separation", "set_boundary_separation", "(Oi)",
                       obj(mtx), boundary_separation);
}

AMGX_RC AMGX_matrix_comm_from_maps(AMGX_matrix_handle, int, int, int,
                                   const int *, const int *, const int *,
                                   const int *, const int *) {
    // map-based comm wiring needs a live multi-process torch.distributed
    // context; C hosts use upload_all_global/upload_distributed instead
    return AMGX_RC_NOT_IMPLEMENTED;
}
AMGX_RC AMGX_matrix_comm_from_maps_one_ring(AMGX_matrix_handle, int, int,
                                            const int *, const int *,
                                            const int **, const int *,
                                            const int **) {
    return AMGX_RC_NOT_IMPLEMENTED;
}

AMGX_RC AMGX_matrix_attach_coloring(AMGX_matrix_handle mtx, int *row_coloring,
                                    int num_rows, int num_colors) {
    Gil gil;
    PyObject *arr = np_from_mem(row_coloring, num_rows, "int32", 4);
    if (!arr) return AMGX_RC_INTERNAL;
    return unpack_rc(call_capi("AMGX_matrix_attach_coloring",
                               Py_BuildValue("(ONi)", obj(mtx), arr,
                                             num_colors)),
                     nullptr, 0, "attach_coloring");
}

AMGX_RC AMGX_matrix_attach_geometry(AMGX_matrix_handle mtx, double *geox,
                                    double *geoy, double *geoz, int n) {
    Gil gil;
    PyObject *gx = np_from_mem(geox, n, "float64", 8);
    PyObject *gy = np_from_mem(geoy, n, "float64", 8);
    PyObject *gz = geoz ? np_from_mem(geoz, n, "float64", 8)
                        : (Py_INCREF(Py_None), Py_None);
    if (!gx || !gy || !gz) {
        Py_XDECREF(gx); Py_XDECREF(gy); Py_XDECREF(gz);
        return AMGX_RC_INTERNAL;
    }
    return unpack_rc(call_capi("AMGX_matrix_attach_geometry",
                               Py_BuildValue("(ONNNi)", obj(mtx), gx, gy, gz,
                                             n)),
                     nullptr, 0, "attach_geometry");
}

AMGX_RC AMGX_matrix_check_symmetry(AMGX_matrix_handle mtx,
                                   int *structurally_symmetric,
                                   int *symmetric) {
    Gil gil;
    PyObject *out[2] = {nullptr, nullptr};
    AMGX_RC rc = unpack_rc(call_capi("AMGX_matrix_check_symmetry",
                                     Py_BuildValue("(O)", obj(mtx))),
                           out, 2, "check_symmetry");
    if (rc == AMGX_RC_OK) {
        if (structurally_symmetric)
            *structurally_symmetric = PyObject_IsTrue(out[0]);
        if (symmetric) *symmetric = PyObject_IsTrue(out[1]);
    }
    Py_XDECREF(out[0]); Py_XDECREF(out[1]);
    return rc;
}

AMGX_RC AMGX_matrix_check_diag_dominant(AMGX_matrix_handle mtx,
                                        int *diag_dominant) {
    Gil gil;
    PyObject *out = nullptr;
    AMGX_RC rc = unpack_rc(call_capi("AMGX_matrix_check_diag_dominant",
                                     Py_BuildValue("(O)", obj(mtx))),
                           &out, 1, "check_diag_dominant");
    if (rc == AMGX_RC_OK && diag_dominant)
        *diag_dominant = PyObject_IsTrue(out);
    Py_XDECREF(out);
    return rc;
}

static AMGX_RC upload_global_common(const char *fn, AMGX_matrix_handle mtx,
                                    int n_global, int n, int nnz,
                                    int block_dimx, int block_dimy,
                                    const int *row_ptrs,
                                    const void *col_indices_global,
                                    const void *data, const void *diag_data,
                                    int allocated_halo_depth,
                                    int num_import_rings,
                                    const int *partition_vector,
                                    bool cols32) {
    Gil gil;
    AMGX_Mode mode = handle_mode(obj(mtx));
    size_t es = mat_elem_size(mode);
    const char *dt = mat_dtype(mode);
    long bb = (long)block_dimx * block_dimy;
    PyObject *ro = np_from_mem(row_ptrs, n + 1, "int32", 4);
    PyObject *ci = cols32 ? np_from_mem(col_indices_global, nnz, "int32", 4)
                          : np_from_mem(col_indices_global, nnz, "int64", 8);
    PyObject *va = np_from_mem(data, (Py_ssize_t)nnz * bb, dt, es);
    PyObject *dg = diag_data
                       ? np_from_mem(diag_data, (Py_ssize_t)n * bb, dt, es)
                       : (Py_INCREF(Py_None), Py_None);
    PyObject *pv = partition_vector
                       ? np_from_mem(partition_vector, n_global, "int32", 4)
                       : (Py_INCREF(Py_None), Py_None);
    if (!ro || !ci || !va || !dg || !pv) {
        Py_XDECREF(ro); Py_XDECREF(ci); Py_XDECREF(va); Py_XDECREF(dg);
        Py_XDECREF(pv);
        return AMGX_RC_INTERNAL;
    }
    return unpack_rc(call_capi(fn,
                               Py_BuildValue("(OiiiiiNNNNiiN)", obj(mtx),
                                             n_global, n, nnz, block_dimx,
                                             block_dimy, ro, ci, va, dg,
                                             allocated_halo_depth,
                                             num_import_rings, pv)),
                     nullptr, 0, fn);
}

AMGX_RC AMGX_matrix_upload_all_global(AMGX_matrix_handle mtx, int n_global,
                                      int n, int nnz, int block_dimx,
                                      int block_dimy, const int *row_ptrs,
                                      const void *col_indices_global,
                                      const void *data, const void *diag_data,
                                      int allocated_halo_depth,
                                      int num_import_rings,
                                      const int *partition_vector) {
    return upload_global_common("AMGX_matrix_upload_all_global", mtx,
                                n_global, n, nnz, block_dimx, block_dimy,
                                row_ptrs, col_indices_global, data, diag_data,
                                allocated_halo_depth, num_import_rings,
                                partition_vector, false);
}

AMGX_RC AMGX_matrix_upload_all_global_32(
    AMGX_matrix_handle mtx, int n_global, int n, int nnz, int block_dimx,
    int block_dimy, const int *row_ptrs, const void *col_indices_global,
    const void *data, const void *diag_data, int allocated_halo_depth,
    int num_import_rings, const int *partition_vector) {
    return upload_global_common("AMGX_matrix_upload_all_global_32", mtx,
                                n_global, n, nnz, block_dimx, block_dimy,
                                row_ptrs, col_indices_global, data, diag_data,
                                allocated_halo_depth, num_import_rings,
                                partition_vector, true);
}

AMGX_RC AMGX_matrix_upload_distributed(
    AMGX_matrix_handle mtx, int n_global, int n, int nnz, int block_dimx,
    int block_dimy, const int *row_ptrs, const void *col_indices_global,
    const void *data, const void *diag_data,
    AMGX_distribution_handle distribution) {
    Gil gil;
    AMGX_Mode mode = handle_mode(obj(mtx));
    size_t es = mat_elem_size(mode);
    const char *dt = mat_dtype(mode);
    long bb = (long)block_dimx * block_dimy;
    // col index width from the distribution handle's 32-bit flag
    int cols32 = 0;
    PyObject *f = PyObject_GetAttrString(obj(distribution), "cols32");
    if (f) { cols32 = PyObject_IsTrue(f); Py_DECREF(f); }
    else PyErr_Clear();
    PyObject *ro = np_from_mem(row_ptrs, n + 1, "int32", 4);
    PyObject *ci = cols32 ? np_from_mem(col_indices_global, nnz, "int32", 4)
                          : np_from_mem(col_indices_global, nnz, "int64", 8);
    PyObject *va = np_from_mem(data, (Py_ssize_t)nnz * bb, dt, es);
    PyObject *dg = diag_data
                       ? np_from_mem(diag_data, (Py_ssize_t)n * bb, dt, es)
                       : (Py_INCREF(Py_None), Py_None);
    if (!ro || !ci || !va || !dg) {
        Py_XDECREF(ro); Py_XDECREF(ci); Py_XDECREF(va); Py_XDECREF(dg);
        return AMGX_RC_INTERNAL;
    }
    return unpack_rc(call_capi("AMGX_matrix_upload_distributed",
                               Py_BuildValue("(OiiiiiNNNNO)", obj(mtx),
                                             n_global, n, nnz, block_dimx,
                                             block_dimy, ro, ci, va, dg,
                                             obj(distribution))),
                     nullptr, 0, "matrix_upload_distributed");
}

/* -------------------------------------------------------------- vector */
AMGX_RC AMGX_vector_create(AMGX_vector_handle *vec, AMGX_resources_handle rsc,
                           AMGX_Mode mode) {
    const char *ms = mode_str(mode);
    if (!ms) return AMGX_RC_BAD_MODE;
    return create_genericv("AMGX_vector_create", (void **)vec, "vector_create",
                          "(Os)", obj(rsc), ms);
}

AMGX_RC AMGX_vector_destroy(AMGX_vector_handle vec) {
    if (!vec) return AMGX_RC_BAD_PARAMETERS;
    Gil gil;
    AMGX_RC rc = unpack_rc(call_capi("AMGX_vector_destroy",
                                     Py_BuildValue("(O)", obj(vec))),
                           nullptr, 0, "vector_destroy");
    Py_DECREF(obj(vec));
    return rc;
}

AMGX_RC AMGX_vector_upload(AMGX_vector_handle vec, int n, int block_dim,
                           const void *data) {
    Gil gil;
    AMGX_Mode mode = handle_mode(obj(vec));
    PyObject *arr = np_from_mem(data, (Py_ssize_t)n * block_dim,
                                vec_dtype(mode), vec_elem_size(mode));
    if (!arr) return AMGX_RC_INTERNAL;
    return unpack_rc(call_capi("AMGX_vector_upload",
                               Py_BuildValue("(OiiN)", obj(vec), n, block_dim,
                                             arr)),
                     nullptr, 0, "vector_upload");
}

AMGX_RC AMGX_vector_set_zero(AMGX_vector_handle vec, int n, int block_dim) {
    return simple_callv("AMGX_vector_set_zero", "vector_set_zero", "(Oii)",
                       obj(vec), n, block_dim);
}

AMGX_RC AMGX_vector_set_random(AMGX_vector_handle vec, int n) {
    return simple_callv("AMGX_vector_set_random", "vector_set_random", "(Oi)",
                       obj(vec), n);
}

AMGX_RC AMGX_vector_download(const AMGX_vector_handle vec, void *data) {
    Gil gil;
    PyObject *out = nullptr;
    AMGX_RC rc = unpack_rc(call_capi("AMGX_vector_download",
                                     Py_BuildValue("(O)", obj(vec))),
                           &out, 1, "vector_download");
    if (rc == AMGX_RC_OK && !copy_out(out, data)) rc = AMGX_RC_INTERNAL;
    Py_XDECREF(out);
    return rc;
}

AMGX_RC AMGX_vector_get_size(const AMGX_vector_handle vec, int *n,
                             int *block_dim) {
    Gil gil;
    PyObject *out[2] = {nullptr, nullptr};
    AMGX_RC rc = unpack_rc(call_capi("AMGX_vector_get_size",
                                     Py_BuildValue("(O)", obj(vec))),
                           out, 2, "vector_get_size");
    if (rc == AMGX_RC_OK) {
        if (n) *n = (int)PyLong_AsLong(out[0]);
        if (block_dim) *block_dim = (int)PyLong_AsLong(out[1]);
    }
    Py_XDECREF(out[0]); Py_XDECREF(out[1]);
    return rc;
}

AMGX_RC AMGX_vector_bind(AMGX_vector_handle vec,
                         const AMGX_matrix_handle mtx) {
    return simple_callv("AMGX_vector_bind", "vector_bind", "(OO)",
                       obj(vec), obj(mtx));
}

/* -------------------------------------------------------------- solver */
AMGX_RC AMGX_solver_create(AMGX_solver_handle *slv, AMGX_resources_handle rsc,
                           AMGX_Mode mode,
                           const AMGX_config_handle cfg_solver) {
    const char *ms = mode_str(mode);
    if (!ms) return AMGX_RC_BAD_MODE;
    return create_genericv("AMGX_solver_create", (void **)slv, "solver_create",
                          "(OsO)", obj(rsc), ms,
                                        obj(cfg_solver));
}

AMGX_RC AMGX_solver_destroy(AMGX_solver_handle slv) {
    if (!slv) return AMGX_RC_BAD_PARAMETERS;
    Gil gil;
    AMGX_RC rc = unpack_rc(call_capi("AMGX_solver_destroy",
                                     Py_BuildValue("(O)", obj(slv))),
                           nullptr, 0, "solver_destroy");
    Py_DECREF(obj(slv));
    return rc;
}

AMGX_RC AMGX_solver_setup(AMGX_solver_handle slv, AMGX_matrix_handle mtx) {
    return simple_callv("AMGX_solver_setup", "solver_setup", "(OO)",
                       obj(slv), obj(mtx));
}

AMGX_RC AMGX_solver_resetup(AMGX_solver_handle slv, AMGX_matrix_handle mtx) {
    return simple_callv("AMGX_solver_resetup", "solver_resetup", "(OO)",
                       obj(slv), obj(mtx));
}

AMGX_RC AMGX_solver_solve(AMGX_solver_handle slv, AMGX_vector_handle rhs,
                          AMGX_vector_handle sol) {
    return simple_callv("AMGX_solver_solve", "solver_solve", "(OOO)",
                       obj(slv), obj(rhs), obj(sol));
}

AMGX_RC AMGX_solver_solve_with_0_initial_guess(AMGX_solver_handle slv,
                                               AMGX_vector_handle rhs,
                                               AMGX_vector_handle sol) {
    return simple_callv("AMGX_solver_solve_with_0_initial_guess", "solver_solve0", "(OOO)",
                       obj(slv), obj(rhs), obj(sol));
}

AMGX_RC AMGX_solver_get_iterations_number(AMGX_solver_handle slv, int *n) {
    Gil gil;
    PyObject *out = nullptr;
    AMGX_RC rc = unpack_rc(call_capi("AMGX_solver_get_iterations_number",
                                     Py_BuildValue("(O)", obj(slv))),
                           &out, 1, "get_iterations_number");
    if (rc == AMGX_RC_OK && n) *n = (int)PyLong_AsLong(out);
    Py_XDECREF(out);
    return rc;
}

AMGX_RC AMGX_solver_get_iteration_residual(AMGX_solver_handle slv, int it,
                                           int idx, double *res) {
    Gil gil;
    PyObject *out = nullptr;
    AMGX_RC rc = unpack_rc(call_capi("AMGX_solver_get_iteration_residual",
                                     Py_BuildValue("(Oii)", obj(slv), it,
                                                   idx)),
                           &out, 1, "get_iteration_residual");
    if (rc == AMGX_RC_OK && res) *res = PyFloat_AsDouble(out);
    Py_XDECREF(out);
    return rc;
}

AMGX_RC AMGX_solver_get_status(AMGX_solver_handle slv,
                               AMGX_SOLVE_STATUS *st) {
    Gil gil;
    PyObject *out = nullptr;
    AMGX_RC rc = unpack_rc(call_capi("AMGX_solver_get_status",
                                     Py_BuildValue("(O)", obj(slv))),
                           &out, 1, "get_status");
    if (rc == AMGX_RC_OK && st)
        *st = (AMGX_SOLVE_STATUS)PyLong_AsLong(out);
    Py_XDECREF(out);
    return rc;
}

AMGX_RC AMGX_solver_calculate_residual_norm(AMGX_solver_handle solver,
                                            AMGX_matrix_handle mtx,
                                            AMGX_vector_handle rhs,
                                            AMGX_vector_handle x,
                                            void *norm_vector) {
    Gil gil;
    PyObject *out = nullptr;
    AMGX_RC rc = unpack_rc(call_capi("AMGX_solver_calculate_residual_norm",
                                     Py_BuildValue("(OOOO)", obj(solver),
                                                   obj(mtx), obj(rhs),
                                                   obj(x))),
                           &out, 1, "calculate_residual_norm");
    if (rc == AMGX_RC_OK && norm_vector) {
        if (PyFloat_Check(out) || PyLong_Check(out))
            *(double *)norm_vector = PyFloat_AsDouble(out);
        else if (!copy_out(out, norm_vector))
            rc = AMGX_RC_INTERNAL;
    }
    Py_XDECREF(out);
    return rc;
}

AMGX_RC AMGX_solver_register_print_callback(AMGX_print_callback func) {
    return AMGX_register_print_callback(func);
}

/* ------------------------------------------------------------ system IO */
AMGX_RC AMGX_write_system(const AMGX_matrix_handle mtx,
                          const AMGX_vector_handle rhs,
                          const AMGX_vector_handle sol,
                          const char *filename) {
    return simple_callv("AMGX_write_system", "write_system", "(OOOs)",
                       obj(mtx),
                                     rhs ? obj(rhs) : Py_None,
                                     sol ? obj(sol) : Py_None, filename);
}

AMGX_RC AMGX_write_system_distributed(
    const AMGX_matrix_handle mtx, const AMGX_vector_handle rhs,
    const AMGX_vector_handle sol, const char *filename, int, int,
    const int *, int, const int *) {
    return simple_callv("AMGX_write_system_distributed", "write_system_distributed", "(OOOs)",
                       obj(mtx),
                                     rhs ? obj(rhs) : Py_None,
                                     sol ? obj(sol) : Py_None, filename);
}

AMGX_RC AMGX_read_system(AMGX_matrix_handle mtx, AMGX_vector_handle rhs,
                         AMGX_vector_handle sol, const char *filename) {
    return simple_callv("AMGX_read_system", "read_system", "(OOOs)",
                       obj(mtx),
                                     rhs ? obj(rhs) : Py_None,
                                     sol ? obj(sol) : Py_None, filename);
}

AMGX_RC AMGX_read_system_distributed(AMGX_matrix_handle mtx,
                                     AMGX_vector_handle rhs,
                                     AMGX_vector_handle sol,
                                     const char *filename,
                                     int allocated_halo_depth,
                                     int num_partitions,
                                     const int *partition_sizes,
                                     int partition_vector_size,
                                     const int *partition_vector) {
    Gil gil;
    (void)partition_sizes;
    PyObject *pv = partition_vector
                       ? np_from_mem(partition_vector, partition_vector_size,
                                     "int32", 4)
                       : (Py_INCREF(Py_None), Py_None);
    if (!pv) return AMGX_RC_INTERNAL;
    return unpack_rc(call_capi("AMGX_read_system_distributed",
                               Py_BuildValue("(OOOsiiN)", obj(mtx),
                                             rhs ? obj(rhs) : Py_None,
                                             sol ? obj(sol) : Py_None,
                                             filename, allocated_halo_depth,
                                             num_partitions, pv)),
                     nullptr, 0, "read_system_distributed");
}

AMGX_RC AMGX_read_system_global(AMGX_matrix_handle mtx,
                                AMGX_vector_handle rhs,
                                AMGX_vector_handle sol, const char *filename,
                                int allocated_halo_depth,
                                int num_import_rings,
                                int partition_vector_size,
                                const int *partition_vector) {
    Gil gil;
    PyObject *pv = partition_vector
                       ? np_from_mem(partition_vector, partition_vector_size,
                                     "int32", 4)
                       : (Py_INCREF(Py_None), Py_None);
    if (!pv) return AMGX_RC_INTERNAL;
    return unpack_rc(call_capi("AMGX_read_system_global",
                               Py_BuildValue("(OOOsiiN)", obj(mtx),
                                             rhs ? obj(rhs) : Py_None,
                                             sol ? obj(sol) : Py_None,
                                             filename, allocated_halo_depth,
                                             num_import_rings, pv)),
                     nullptr, 0, "read_system_global");
}

AMGX_RC AMGX_read_system_maps_one_ring(
    int *, int *, int *, int *, int **, int **, void **, void **, void **,
    void **, int *, int **, int **, int ***, int **, int ***,
    AMGX_resources_handle, AMGX_Mode, const char *, int, int, const int *,
    int, const int *) {
    return AMGX_RC_NOT_IMPLEMENTED;   // C hosts use read_system_global
}

AMGX_RC AMGX_free_system_maps_one_ring(int *, int *, void *, void *, void *,
                                       void *, int, int *, int *, int **,
                                       int *, int **) {
    return AMGX_RC_NOT_IMPLEMENTED;
}

/* ------------------------------------------------------------ utilities */
AMGX_RC AMGX_generate_distributed_poisson_7pt(
    AMGX_matrix_handle mtx, AMGX_vector_handle rhs, AMGX_vector_handle sol,
    int allocated_halo_depth, int num_import_rings, int nx, int ny, int nz,
    int px, int py, int pz) {
    return simple_callv("AMGX_generate_distributed_poisson_7pt", "generate_distributed_poisson_7pt", "(OOOiiiiiiii)",
                       obj(mtx), obj(rhs),
                                     obj(sol), allocated_halo_depth,
                                     num_import_rings, nx, ny, nz, px, py,
                                     pz);
}

/* ---------------------------------------------------------- eigensolvers */
AMGX_RC AMGX_eigensolver_create(AMGX_eigensolver_handle *ret,
                                AMGX_resources_handle rsc, AMGX_Mode mode,
                                const AMGX_config_handle cfg) {
    const char *ms = mode_str(mode);
    if (!ms) return AMGX_RC_BAD_MODE;
    return create_genericv("AMGX_eigensolver_create", (void **)ret,
                           "eigensolver_create", "(OsO)", obj(rsc), ms,
                           obj(cfg));
}

AMGX_RC AMGX_eigensolver_setup(AMGX_eigensolver_handle s,
                               AMGX_matrix_handle mtx) {
    return simple_callv("AMGX_eigensolver_setup", "eigensolver_setup",
                        "(OO)", obj(s), obj(mtx));
}

AMGX_RC AMGX_eigensolver_pagerank_setup(AMGX_eigensolver_handle s,
                                        AMGX_vector_handle a) {
    return simple_callv("AMGX_eigensolver_pagerank_setup",
                        "eigensolver_pagerank_setup", "(OO)", obj(s),
                        obj(a));
}

AMGX_RC AMGX_eigensolver_solve(AMGX_eigensolver_handle s,
                               AMGX_vector_handle x) {
    return simple_callv("AMGX_eigensolver_solve", "eigensolver_solve",
                        "(OO)", obj(s), x ? obj(x) : Py_None);
}

AMGX_RC AMGX_eigensolver_destroy(AMGX_eigensolver_handle s) {
    if (!s) return AMGX_RC_BAD_PARAMETERS;
    Gil gil;
    AMGX_RC rc = unpack_rc(call_capi("AMGX_eigensolver_destroy",
                                     Py_BuildValue("(O)", obj(s))),
                           nullptr, 0, "eigensolver_destroy");
    Py_DECREF(obj(s));
    return rc;
}

AMGX_RC AMGX_write_parameters_description(char *filename,
                                          AMGX_GET_PARAMS_DESC_FLAG mode) {
    (void)mode;   // JSON/file is the one persisted form here
    return simple_callv("AMGX_write_parameters_description", "write_parameters_description", "(s)",
                       filename);
}

}  // extern "C"
