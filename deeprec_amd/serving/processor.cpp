// processor.cpp — the serving C ABI (≙ reference
// serving/processor/serving/processor.cc:8-102: initialize / process /
// batch_process, pluggable into EAS or any RPC shell via dlopen).
//
// MI355X-native design: the ABI shell embeds CPython and drives the
// engine's Predictor (deeprec_amd.serving.c_entry); the compute under it
// is the same HIP/torch stack training uses. Payloads are length-tagged
// byte buffers (JSON in the default entry, matching the reference's
// request/response proto role); output buffers are malloc'd and released
// with free_buffer.

#define PY_SSIZE_T_CLEAN
#include <Python.h>

#include <cstdlib>
#include <cstring>

namespace {

struct ModelBuf {
  PyObject* handle;  // the Python-side predictor handle
};

PyObject* entry_module() {
  PyObject* mod = PyImport_ImportModule("deeprec_amd.serving.c_entry");
  return mod;  // nullptr on failure (exception set)
}

int fill_output(PyObject* bytes, void** output_data, int* output_size) {
  char* buf = nullptr;
  Py_ssize_t len = 0;
  if (PyBytes_AsStringAndSize(bytes, &buf, &len) != 0) return -3;
  void* out = std::malloc(len);
  if (!out) return -4;
  std::memcpy(out, buf, len);
  *output_data = out;
  *output_size = (int)len;
  return 0;
}

}  // namespace

extern "C" {

// state: 0 ok, negative = error code. Returns an opaque model buffer.
void* initialize(const char* model_entry, const char* model_config,
                 int* state) {
  if (!Py_IsInitialized()) {
    Py_InitializeEx(0);
    // release the GIL acquired by initialization so worker threads (and
    // this thread's PyGILState_Ensure below) manage it uniformly
    PyEval_SaveThread();
  }
  PyGILState_STATE g = PyGILState_Ensure();
  ModelBuf* mb = nullptr;
  PyObject* mod = entry_module();
  if (mod) {
    PyObject* fn = PyObject_GetAttrString(mod, "initialize");
    PyObject* res =
        fn ? PyObject_CallFunction(fn, "ss", model_entry ? model_entry : "",
                                   model_config ? model_config : "{}")
           : nullptr;
    if (res) {
      mb = (ModelBuf*)std::malloc(sizeof(ModelBuf));
      mb->handle = res;  // owned reference
      if (state) *state = 0;
    } else {
      PyErr_Print();
      if (state) *state = -1;
    }
    Py_XDECREF(fn);
    Py_DECREF(mod);
  } else {
    PyErr_Print();
    if (state) *state = -2;
  }
  PyGILState_Release(g);
  return mb;
}

int process(void* model_buf, const void* input_data, int input_size,
            void** output_data, int* output_size) {
  if (!model_buf || !input_data || !output_data || !output_size) return -1;
  ModelBuf* mb = (ModelBuf*)model_buf;
  PyGILState_STATE g = PyGILState_Ensure();
  int rc = -1;
  PyObject* mod = entry_module();
  if (mod) {
    PyObject* fn = PyObject_GetAttrString(mod, "process");
    PyObject* res = fn ? PyObject_CallFunction(fn, "Oy#", mb->handle,
                                               (const char*)input_data,
                                               (Py_ssize_t)input_size)
                       : nullptr;
    if (res && PyBytes_Check(res)) {
      rc = fill_output(res, output_data, output_size);
    } else if (!res) {
      PyErr_Print();
      rc = -2;
    }
    Py_XDECREF(res);
    Py_XDECREF(fn);
    Py_DECREF(mod);
  }
  PyGILState_Release(g);
  return rc;
}

// inputs: `num` length-tagged buffers; one response buffer (a JSON array
// in the default entry).
int batch_process(void* model_buf, const void** input_datas,
                  const int* input_sizes, int num, void** output_data,
                  int* output_size) {
  if (!model_buf || num < 0) return -1;
  ModelBuf* mb = (ModelBuf*)model_buf;
  PyGILState_STATE g = PyGILState_Ensure();
  int rc = -1;
  PyObject* mod = entry_module();
  if (mod) {
    PyObject* lst = PyList_New(num);
    for (int i = 0; i < num; ++i) {
      PyList_SetItem(lst, i,
                     PyBytes_FromStringAndSize((const char*)input_datas[i],
                                               input_sizes[i]));
    }
    PyObject* fn = PyObject_GetAttrString(mod, "batch_process");
    PyObject* res =
        fn ? PyObject_CallFunctionObjArgs(fn, mb->handle, lst, nullptr)
           : nullptr;
    if (res && PyBytes_Check(res)) {
      rc = fill_output(res, output_data, output_size);
    } else if (!res) {
      PyErr_Print();
      rc = -2;
    }
    Py_XDECREF(res);
    Py_XDECREF(fn);
    Py_DECREF(lst);
    Py_DECREF(mod);
  }
  PyGILState_Release(g);
  return rc;
}

void free_buffer(void* p) { std::free(p); }

void shutdown_processor(void* model_buf) {
  if (!model_buf) return;
  ModelBuf* mb = (ModelBuf*)model_buf;
  PyGILState_STATE g = PyGILState_Ensure();
  PyObject* mod = entry_module();
  if (mod) {
    PyObject* fn = PyObject_GetAttrString(mod, "shutdown");
    if (fn) {
      PyObject* r = PyObject_CallFunctionObjArgs(fn, mb->handle, nullptr);
      Py_XDECREF(r);
      Py_DECREF(fn);
    }
    Py_DECREF(mod);
  }
  Py_DECREF(mb->handle);
  PyGILState_Release(g);
  std::free(mb);
}

}  // extern "C"
