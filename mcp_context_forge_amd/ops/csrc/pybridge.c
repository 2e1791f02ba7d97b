/* CPython bridge for the response assembly hot loops.
 *
 * The GPU pipeline's last host step turns (arena bytes, per-row spans) into
 * per-request PyBytes in the responses list. In Python that loop costs
 * ~0.4 us/row (≈3 ms per 8k batch — the single largest host stage, see
 * profiles/README.md). Here it is one C loop: PyBytes_FromStringAndSize
 * straight out of the numpy arena buffer, PyList_SetItem (steals the ref,
 * frees any prior entry).
 *
 * Plain CPython C API extension (no hipcc, no torch ABI): compiled by
 * ops/build.py with the system compiler into forge_pybridge.so in-tree.
 */
#define PY_SSIZE_T_CLEAN
#include <Python.h>
#include <stdint.h>

/* scatter_slices(arena, rb, re, dest, out_list [, prefix, suffix]) -> count
 *
 * arena: buffer (numpy uint8 or bytes) with the spliced responses
 * rb/re: int64 buffers, span of row k (rb[k] < 0 -> skip row)
 * dest:  int64 buffer, index into out_list for row k
 * out_list: Python list to scatter into
 * prefix/suffix: optional bytes wrapped around each slice (0-copy concat)
 */
static PyObject *scatter_slices(PyObject *self, PyObject *args) {
    Py_buffer arena, rb, re, dest;
    PyObject *out_list;
    const char *prefix = NULL, *suffix = NULL;
    Py_ssize_t prefix_len = 0, suffix_len = 0;

    if (!PyArg_ParseTuple(args, "y*y*y*y*O!|y#y#", &arena, &rb, &re, &dest,
                          &PyList_Type, &out_list,
                          &prefix, &prefix_len, &suffix, &suffix_len))
        return NULL;

    const char *base = (const char *)arena.buf;
    const int64_t *rbp = (const int64_t *)rb.buf;
    const int64_t *rep = (const int64_t *)re.buf;
    const int64_t *dp = (const int64_t *)dest.buf;
    Py_ssize_t n = (Py_ssize_t)(rb.len / (Py_ssize_t)sizeof(int64_t));
    Py_ssize_t list_len = PyList_GET_SIZE(out_list);
    Py_ssize_t arena_len = arena.len;
    Py_ssize_t count = 0;

    for (Py_ssize_t k = 0; k < n; k++) {
        int64_t b = rbp[k], e = rep[k], d = dp[k];
        if (b < 0 || e < b || d < 0 || d >= list_len || e > arena_len)
            continue;
        PyObject *obj;
        if (prefix_len || suffix_len) {
            Py_ssize_t total = prefix_len + (Py_ssize_t)(e - b) + suffix_len;
            obj = PyBytes_FromStringAndSize(NULL, total);
            if (obj) {
                char *w = PyBytes_AS_STRING(obj);
                memcpy(w, prefix, (size_t)prefix_len);
                memcpy(w + prefix_len, base + b, (size_t)(e - b));
                memcpy(w + prefix_len + (e - b), suffix, (size_t)suffix_len);
            }
        } else {
            obj = PyBytes_FromStringAndSize(base + b, (Py_ssize_t)(e - b));
        }
        if (!obj) {
            PyBuffer_Release(&arena); PyBuffer_Release(&rb);
            PyBuffer_Release(&re); PyBuffer_Release(&dest);
            return NULL;
        }
        PyList_SetItem(out_list, d, obj); /* steals ref, frees old entry */
        count++;
    }
    PyBuffer_Release(&arena); PyBuffer_Release(&rb);
    PyBuffer_Release(&re); PyBuffer_Release(&dest);
    return PyLong_FromSsize_t(count);
}

/* slices_list(arena, beg, end) -> list[bytes|None]
 * Bulk-extract spans into a fresh list (None where beg<0). */
static PyObject *slices_list(PyObject *self, PyObject *args) {
    Py_buffer arena, beg, end;
    if (!PyArg_ParseTuple(args, "y*y*y*", &arena, &beg, &end))
        return NULL;
    const char *base = (const char *)arena.buf;
    const int64_t *bp = (const int64_t *)beg.buf;
    const int64_t *ep = (const int64_t *)end.buf;
    Py_ssize_t n = (Py_ssize_t)(beg.len / (Py_ssize_t)sizeof(int64_t));
    Py_ssize_t arena_len = arena.len;
    PyObject *out = PyList_New(n);
    if (!out) goto fail;
    for (Py_ssize_t k = 0; k < n; k++) {
        int64_t b = bp[k], e = ep[k];
        PyObject *obj;
        if (b < 0 || e < b || e > arena_len) {
            Py_INCREF(Py_None);
            obj = Py_None;
        } else {
            obj = PyBytes_FromStringAndSize(base + b, (Py_ssize_t)(e - b));
            if (!obj) { Py_DECREF(out); goto fail; }
        }
        PyList_SET_ITEM(out, k, obj);
    }
    PyBuffer_Release(&arena); PyBuffer_Release(&beg); PyBuffer_Release(&end);
    return out;
fail:
    PyBuffer_Release(&arena); PyBuffer_Release(&beg); PyBuffer_Release(&end);
    return NULL;
}

/* concat_with_offsets(list_of_bytes) -> (bytes, int64 offsets bytearray)
 * One C pass: total size, offsets array (n+1 int64 little-endian in a
 * bytes object the caller views as numpy), and the joined blob. */
static PyObject *concat_with_offsets(PyObject *self, PyObject *args) {
    PyObject *lst;
    if (!PyArg_ParseTuple(args, "O!", &PyList_Type, &lst))
        return NULL;
    Py_ssize_t n = PyList_GET_SIZE(lst);
    PyObject *offs_obj = PyBytes_FromStringAndSize(NULL, (n + 1) * (Py_ssize_t)sizeof(int64_t));
    if (!offs_obj) return NULL;
    int64_t *offs = (int64_t *)PyBytes_AS_STRING(offs_obj);
    int64_t total = 0;
    offs[0] = 0;
    for (Py_ssize_t i = 0; i < n; i++) {
        PyObject *it = PyList_GET_ITEM(lst, i);
        Py_ssize_t sz;
        if (PyBytes_Check(it)) sz = PyBytes_GET_SIZE(it);
        else { PyErr_SetString(PyExc_TypeError, "expected bytes"); Py_DECREF(offs_obj); return NULL; }
        total += sz;
        offs[i + 1] = total;
    }
    PyObject *blob = PyBytes_FromStringAndSize(NULL, total);
    if (!blob) { Py_DECREF(offs_obj); return NULL; }
    char *w = PyBytes_AS_STRING(blob);
    for (Py_ssize_t i = 0; i < n; i++) {
        PyObject *it = PyList_GET_ITEM(lst, i);
        Py_ssize_t sz = PyBytes_GET_SIZE(it);
        memcpy(w + offs[i], PyBytes_AS_STRING(it), (size_t)sz);
    }
    return Py_BuildValue("NN", blob, offs_obj);
}

/* pack_frame(ids_int64_buffer, list_of_bytes_or_None) -> bytes
 * Owner->worker edge frame: [u32 payload_bytes][u32 n] payload
 * n x {[u64 id][u32 len][bytes]}; None -> len = 0xFFFFFFFF, no body. */
static PyObject *pack_frame(PyObject *self, PyObject *args) {
    Py_buffer ids;
    PyObject *outs;
    if (!PyArg_ParseTuple(args, "y*O!", &ids, &PyList_Type, &outs))
        return NULL;
    Py_ssize_t n = (Py_ssize_t)(ids.len / (Py_ssize_t)sizeof(int64_t));
    if (PyList_GET_SIZE(outs) != n) {
        PyBuffer_Release(&ids);
        PyErr_SetString(PyExc_ValueError, "ids/outs length mismatch");
        return NULL;
    }
    const int64_t *idp = (const int64_t *)ids.buf;
    int64_t payload = 0;
    for (Py_ssize_t i = 0; i < n; i++) {
        PyObject *o = PyList_GET_ITEM(outs, i);
        payload += 12;
        if (o != Py_None) {
            if (!PyBytes_Check(o)) {
                PyBuffer_Release(&ids);
                PyErr_SetString(PyExc_TypeError, "outs must be bytes or None");
                return NULL;
            }
            payload += PyBytes_GET_SIZE(o);
        }
    }
    PyObject *frame = PyBytes_FromStringAndSize(NULL, 8 + payload);
    if (!frame) { PyBuffer_Release(&ids); return NULL; }
    unsigned char *w = (unsigned char *)PyBytes_AS_STRING(frame);
    uint32_t pb = (uint32_t)payload, nn = (uint32_t)n;
    memcpy(w, &pb, 4); memcpy(w + 4, &nn, 4);
    w += 8;
    for (Py_ssize_t i = 0; i < n; i++) {
        PyObject *o = PyList_GET_ITEM(outs, i);
        uint64_t id = (uint64_t)idp[i];
        memcpy(w, &id, 8); w += 8;
        if (o == Py_None) {
            uint32_t nr = 0xFFFFFFFFu;
            memcpy(w, &nr, 4); w += 4;
        } else {
            uint32_t ln = (uint32_t)PyBytes_GET_SIZE(o);
            memcpy(w, &ln, 4); w += 4;
            memcpy(w, PyBytes_AS_STRING(o), ln); w += ln;
        }
    }
    PyBuffer_Release(&ids);
    return frame;
}

/* unpack_frame(payload, n) -> (bytes ids_int64, list_of_bytes)
 * Worker->owner edge frame payload: n x {[u64 id][u32 len][bytes]}. */
static PyObject *unpack_frame(PyObject *self, PyObject *args) {
    Py_buffer pay;
    Py_ssize_t n;
    if (!PyArg_ParseTuple(args, "y*n", &pay, &n))
        return NULL;
    const unsigned char *p = (const unsigned char *)pay.buf;
    Py_ssize_t remain = pay.len;
    PyObject *ids = PyBytes_FromStringAndSize(NULL, n * (Py_ssize_t)sizeof(int64_t));
    PyObject *lst = PyList_New(n);
    if (!ids || !lst) goto fail;
    int64_t *idp = (int64_t *)PyBytes_AS_STRING(ids);
    for (Py_ssize_t i = 0; i < n; i++) {
        if (remain < 12) { PyErr_SetString(PyExc_ValueError, "truncated frame"); goto fail; }
        uint64_t id; uint32_t ln;
        memcpy(&id, p, 8); memcpy(&ln, p + 8, 4);
        p += 12; remain -= 12;
        if ((Py_ssize_t)ln > remain) { PyErr_SetString(PyExc_ValueError, "truncated body"); goto fail; }
        idp[i] = (int64_t)id;
        PyObject *b = PyBytes_FromStringAndSize((const char *)p, (Py_ssize_t)ln);
        if (!b) goto fail;
        PyList_SET_ITEM(lst, i, b);
        p += ln; remain -= ln;
    }
    PyBuffer_Release(&pay);
    return Py_BuildValue("NN", ids, lst);
fail:
    Py_XDECREF(ids); Py_XDECREF(lst);
    PyBuffer_Release(&pay);
    return NULL;
}

static PyMethodDef Methods[] = {
    {"scatter_slices", scatter_slices, METH_VARARGS,
     "scatter arena spans into a responses list as bytes"},
    {"slices_list", slices_list, METH_VARARGS,
     "extract arena spans into a new list of bytes/None"},
    {"concat_with_offsets", concat_with_offsets, METH_VARARGS,
     "join a list of bytes into one blob + int64 offsets"},
    {"pack_frame", pack_frame, METH_VARARGS,
     "pack an owner->worker edge response frame"},
    {"unpack_frame", unpack_frame, METH_VARARGS,
     "unpack a worker->owner edge request frame payload"},
    {NULL, NULL, 0, NULL},
};

static struct PyModuleDef moduledef = {
    PyModuleDef_HEAD_INIT, "forge_pybridge", NULL, -1, Methods,
};

PyMODINIT_FUNC PyInit_forge_pybridge(void) {
    return PyModule_Create(&moduledef);
}
