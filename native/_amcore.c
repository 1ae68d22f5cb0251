/* _amcore: native hot-path primitives for active-monitor-amd.
 *
 * The framework's profile is dominated by copying JSON-shaped Kubernetes
 * objects at every apiserver boundary (see docs/DESIGN.md "Copy discipline").
 * This extension implements the two copy primitives from
 * active_monitor_amd/utils/fastcopy.py in C:
 *
 *   deep_copy(obj)  - recursive private copy of dict/list/tuple trees;
 *                     immutable leaves (str/int/float/bool/None/bytes) are
 *                     shared by refcount; non-JSON objects fall back to
 *                     copy.deepcopy for identical semantics.
 *   snapshot(obj)   - read-optimized object copy: fresh top-level dict with
 *                     "metadata"/"status" deep-copied and all other subtrees
 *                     shared (the store's read-only contract).
 *
 * This is the only native code in the repo by design: the reference
 * (keikoproj/active-monitor) is a pure control-plane controller with no
 * numeric hot path, and its only "native" artifact is the
 * compiled Go binary itself; the equivalent here is compiling the measured
 * hot path of the runtime.
 */
#define PY_SSIZE_T_CLEAN
#include <Python.h>

static PyObject *copy_deepcopy = NULL; /* cached copy.deepcopy */

static PyObject *am_deep_copy_obj(PyObject *obj);

static PyObject *
am_copy_dict(PyObject *obj)
{
    PyObject *out = PyDict_New();
    if (out == NULL)
        return NULL;
    Py_ssize_t pos = 0;
    PyObject *key, *value;
    while (PyDict_Next(obj, &pos, &key, &value)) {
        PyObject *cv = am_deep_copy_obj(value);
        if (cv == NULL) {
            Py_DECREF(out);
            return NULL;
        }
        /* keys in k8s objects are strings (immutable): share by refcount */
        if (PyDict_SetItem(out, key, cv) < 0) {
            Py_DECREF(cv);
            Py_DECREF(out);
            return NULL;
        }
        Py_DECREF(cv);
    }
    return out;
}

static PyObject *
am_copy_list(PyObject *obj)
{
    Py_ssize_t n = PyList_GET_SIZE(obj);
    PyObject *out = PyList_New(n);
    if (out == NULL)
        return NULL;
    for (Py_ssize_t i = 0; i < n; i++) {
        PyObject *cv = am_deep_copy_obj(PyList_GET_ITEM(obj, i));
        if (cv == NULL) {
            Py_DECREF(out);
            return NULL;
        }
        PyList_SET_ITEM(out, i, cv); /* steals */
    }
    return out;
}

static PyObject *
am_copy_tuple(PyObject *obj)
{
    Py_ssize_t n = PyTuple_GET_SIZE(obj);
    PyObject *out = PyTuple_New(n);
    if (out == NULL)
        return NULL;
    for (Py_ssize_t i = 0; i < n; i++) {
        PyObject *cv = am_deep_copy_obj(PyTuple_GET_ITEM(obj, i));
        if (cv == NULL) {
            Py_DECREF(out);
            return NULL;
        }
        PyTuple_SET_ITEM(out, i, cv); /* steals */
    }
    return out;
}

static PyObject *
am_deep_copy_obj(PyObject *obj)
{
    /* immutable leaves: share. Exact-type checks keep subclass semantics on
     * the deepcopy fallback path. */
    if (obj == Py_None || PyUnicode_CheckExact(obj) || PyLong_CheckExact(obj) ||
        PyFloat_CheckExact(obj) || PyBool_Check(obj) || PyBytes_CheckExact(obj)) {
        Py_INCREF(obj);
        return obj;
    }
    if (Py_EnterRecursiveCall(" in _amcore.deep_copy"))
        return NULL;
    PyObject *out;
    if (PyDict_CheckExact(obj))
        out = am_copy_dict(obj);
    else if (PyList_CheckExact(obj))
        out = am_copy_list(obj);
    else if (PyTuple_CheckExact(obj))
        out = am_copy_tuple(obj);
    else
        /* non-JSON payload: identical semantics to the Python fallback */
        out = PyObject_CallFunctionObjArgs(copy_deepcopy, obj, NULL);
    Py_LeaveRecursiveCall();
    return out;
}

static PyObject *
am_deep_copy(PyObject *self, PyObject *obj)
{
    (void)self;
    return am_deep_copy_obj(obj);
}

static PyObject *
am_snapshot(PyObject *self, PyObject *obj)
{
    (void)self;
    if (!PyDict_CheckExact(obj))
        return am_deep_copy_obj(obj);
    PyObject *out = PyDict_Copy(obj);
    if (out == NULL)
        return NULL;
    static const char *private_keys[] = {"metadata", "status", NULL};
    for (int i = 0; private_keys[i] != NULL; i++) {
        PyObject *key = PyUnicode_FromString(private_keys[i]);
        if (key == NULL) {
            Py_DECREF(out);
            return NULL;
        }
        PyObject *value = PyDict_GetItemWithError(out, key);
        if (value == NULL) {
            Py_DECREF(key);
            if (PyErr_Occurred()) {
                Py_DECREF(out);
                return NULL;
            }
            continue;
        }
        PyObject *cv = am_deep_copy_obj(value);
        if (cv == NULL || PyDict_SetItem(out, key, cv) < 0) {
            Py_XDECREF(cv);
            Py_DECREF(key);
            Py_DECREF(out);
            return NULL;
        }
        Py_DECREF(cv);
        Py_DECREF(key);
    }
    return out;
}

static PyMethodDef am_methods[] = {
    {"deep_copy", am_deep_copy, METH_O,
     "Fast deep copy of a JSON-shaped object tree."},
    {"snapshot", am_snapshot, METH_O,
     "Read-optimized object copy: private metadata/status, shared spec."},
    {NULL, NULL, 0, NULL},
};

static struct PyModuleDef am_module = {
    PyModuleDef_HEAD_INIT, "_amcore",
    "Native hot-path primitives for active-monitor-amd.", -1, am_methods,
    NULL, NULL, NULL, NULL,
};

PyMODINIT_FUNC
PyInit__amcore(void)
{
    PyObject *copy_mod = PyImport_ImportModule("copy");
    if (copy_mod == NULL)
        return NULL;
    copy_deepcopy = PyObject_GetAttrString(copy_mod, "deepcopy");
    Py_DECREF(copy_mod);
    if (copy_deepcopy == NULL)
        return NULL;
    return PyModule_Create(&am_module);
}
