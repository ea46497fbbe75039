// Native WAL frame codec (CPU): one-pass frame packing/parsing with a
// slice-by-8 CRC32. The reference implements its segment codec natively
// (crates/arkflow-plugin/src/wal/segment.rs:52-120, wal/crc.rs); this is the
// equivalent for our runtime. Frame layout matches arkflow_amd/wal/store.py:
//   [seq u64 BE | len u32 BE | tag u8 | body | crc32 u32 BE]
// where len = 1 + body bytes and the CRC covers tag+body with the zlib
// (IEEE reflected) polynomial so Python- and native-written logs interop.
#define PY_SSIZE_T_CLEAN
#include <Python.h>

#include <cstdint>
#include <cstring>
#include <vector>

namespace {

uint32_t crc_tab[8][256];

void init_crc_tables() {
  for (uint32_t i = 0; i < 256; i++) {
    uint32_t c = i;
    for (int k = 0; k < 8; k++) c = (c >> 1) ^ (0xEDB88320u & (-(c & 1u)));
    crc_tab[0][i] = c;
  }
  for (uint32_t i = 0; i < 256; i++)
    for (int t = 1; t < 8; t++)
      crc_tab[t][i] =
          (crc_tab[t - 1][i] >> 8) ^ crc_tab[0][crc_tab[t - 1][i] & 0xff];
}

uint32_t crc32_sb8(const uint8_t* p, size_t n, uint32_t crc) {
  crc = ~crc;
  while (n && (reinterpret_cast<uintptr_t>(p) & 7)) {
    crc = (crc >> 8) ^ crc_tab[0][(crc ^ *p++) & 0xff];
    n--;
  }
  while (n >= 8) {
    uint64_t w;
    std::memcpy(&w, p, 8);
    w ^= crc;  // little-endian host: low 4 bytes fold the running crc
    crc = crc_tab[7][w & 0xff] ^ crc_tab[6][(w >> 8) & 0xff] ^
          crc_tab[5][(w >> 16) & 0xff] ^ crc_tab[4][(w >> 24) & 0xff] ^
          crc_tab[3][(w >> 32) & 0xff] ^ crc_tab[2][(w >> 40) & 0xff] ^
          crc_tab[1][(w >> 48) & 0xff] ^ crc_tab[0][(w >> 56) & 0xff];
    p += 8;
    n -= 8;
  }
  while (n--) crc = (crc >> 8) ^ crc_tab[0][(crc ^ *p++) & 0xff];
  return ~crc;
}

inline void put_u64be(uint8_t* p, uint64_t v) {
  for (int i = 7; i >= 0; i--) { p[i] = v & 0xff; v >>= 8; }
}
inline void put_u32be(uint8_t* p, uint32_t v) {
  for (int i = 3; i >= 0; i--) { p[i] = v & 0xff; v >>= 8; }
}
inline uint64_t get_u64be(const uint8_t* p) {
  uint64_t v = 0;
  for (int i = 0; i < 8; i++) v = (v << 8) | p[i];
  return v;
}
inline uint32_t get_u32be(const uint8_t* p) {
  uint32_t v = 0;
  for (int i = 0; i < 4; i++) v = (v << 8) | p[i];
  return v;
}

// encode_frames([(seq, payload_bytes), ...]) -> bytes  (raw 'R' frames)
PyObject* encode_frames(PyObject*, PyObject* args) {
  PyObject* entries;
  if (!PyArg_ParseTuple(args, "O", &entries)) return nullptr;
  PyObject* seq_list = PySequence_Fast(entries, "entries must be a sequence");
  if (!seq_list) return nullptr;
  Py_ssize_t n = PySequence_Fast_GET_SIZE(seq_list);

  std::vector<Py_buffer> bufs(n);
  std::vector<uint64_t> seqs(n);
  size_t total = 0;
  Py_ssize_t got = 0;
  for (Py_ssize_t i = 0; i < n; i++, got++) {
    PyObject* item = PySequence_Fast_GET_ITEM(seq_list, i);
    PyObject* seq_o;
    PyObject* payload_o;
    if (!PyArg_ParseTuple(item, "OO", &seq_o, &payload_o) ||
        PyObject_GetBuffer(payload_o, &bufs[i], PyBUF_SIMPLE) < 0) {
      for (Py_ssize_t j = 0; j < got; j++) PyBuffer_Release(&bufs[j]);
      Py_DECREF(seq_list);
      return nullptr;
    }
    seqs[i] = PyLong_AsUnsignedLongLong(seq_o);
    total += 12 + 1 + (size_t)bufs[i].len + 4;
  }

  PyObject* out = PyBytes_FromStringAndSize(nullptr, (Py_ssize_t)total);
  if (out) {
    uint8_t* w = (uint8_t*)PyBytes_AS_STRING(out);
    for (Py_ssize_t i = 0; i < n; i++) {
      size_t blen = (size_t)bufs[i].len;
      put_u64be(w, seqs[i]);
      put_u32be(w + 8, (uint32_t)(blen + 1));
      w[12] = 'R';
      std::memcpy(w + 13, bufs[i].buf, blen);
      // CRC covers the whole frame incl. the seq/len header (a flipped
      // header byte must fail validation, not replay as a wrong seq)
      put_u32be(w + 13 + blen, crc32_sb8(w, blen + 13, 0));
      w += 13 + blen + 4;
    }
  }
  for (Py_ssize_t i = 0; i < n; i++) PyBuffer_Release(&bufs[i]);
  Py_DECREF(seq_list);
  return out;
}

// encode_frame_parts(seq, [part_buffers...]) -> one raw 'R' frame, packing
// the scatter-gather parts in a single pass (no intermediate join copy)
PyObject* encode_frame_parts(PyObject*, PyObject* args) {
  unsigned long long seq;
  PyObject* parts;
  if (!PyArg_ParseTuple(args, "KO", &seq, &parts)) return nullptr;
  PyObject* fast = PySequence_Fast(parts, "parts must be a sequence");
  if (!fast) return nullptr;
  Py_ssize_t np = PySequence_Fast_GET_SIZE(fast);
  std::vector<Py_buffer> bufs(np);
  size_t body = 0;
  for (Py_ssize_t i = 0; i < np; i++) {
    if (PyObject_GetBuffer(PySequence_Fast_GET_ITEM(fast, i), &bufs[i],
                           PyBUF_SIMPLE) < 0) {
      for (Py_ssize_t j = 0; j < i; j++) PyBuffer_Release(&bufs[j]);
      Py_DECREF(fast);
      return nullptr;
    }
    body += (size_t)bufs[i].len;
  }
  PyObject* out = PyBytes_FromStringAndSize(nullptr,
                                            (Py_ssize_t)(12 + 1 + body + 4));
  if (out) {
    uint8_t* w = (uint8_t*)PyBytes_AS_STRING(out);
    put_u64be(w, seq);
    put_u32be(w + 8, (uint32_t)(body + 1));
    w[12] = 'R';
    uint8_t* p = w + 13;
    for (Py_ssize_t i = 0; i < np; i++) {
      std::memcpy(p, bufs[i].buf, (size_t)bufs[i].len);
      p += bufs[i].len;
    }
    put_u32be(p, crc32_sb8(w, body + 13, 0));
  }
  for (Py_ssize_t i = 0; i < np; i++) PyBuffer_Release(&bufs[i]);
  Py_DECREF(fast);
  return out;
}

// decode_frames(buf) -> list[(seq, tag:int, body_bytes)], torn-tail truncated
PyObject* decode_frames(PyObject*, PyObject* args) {
  Py_buffer buf;
  if (!PyArg_ParseTuple(args, "y*", &buf)) return nullptr;
  const uint8_t* p = (const uint8_t*)buf.buf;
  size_t n = (size_t)buf.len;
  PyObject* out = PyList_New(0);
  size_t pos = 0;
  while (out && pos + 12 <= n) {
    uint64_t seq = get_u64be(p + pos);
    uint32_t ln = get_u32be(p + pos + 8);
    if (pos + 12 + (size_t)ln + 4 > n || ln == 0) break;
    const uint8_t* payload = p + pos + 12;
    if (crc32_sb8(p + pos, ln + 12, 0) != get_u32be(payload + ln)) break;
    PyObject* tup = Py_BuildValue("(KBy#)", (unsigned long long)seq,
                                  (unsigned char)payload[0],
                                  (const char*)payload + 1,
                                  (Py_ssize_t)(ln - 1));
    if (!tup || PyList_Append(out, tup) < 0) {
      Py_XDECREF(tup);
      Py_CLEAR(out);
      break;
    }
    Py_DECREF(tup);
    pos += 12 + ln + 4;
  }
  PyBuffer_Release(&buf);
  return out;
}

PyObject* crc32_py(PyObject*, PyObject* args) {
  Py_buffer buf;
  unsigned int init = 0;
  if (!PyArg_ParseTuple(args, "y*|I", &buf, &init)) return nullptr;
  uint32_t c = crc32_sb8((const uint8_t*)buf.buf, (size_t)buf.len, init);
  PyBuffer_Release(&buf);
  return PyLong_FromUnsignedLong(c);
}

PyMethodDef methods[] = {
    {"encode_frames", encode_frames, METH_VARARGS,
     "encode [(seq, payload)] into raw CRC frames"},
    {"encode_frame_parts", encode_frame_parts, METH_VARARGS,
     "encode one frame from scatter-gather parts (single-pass pack+crc)"},
    {"decode_frames", decode_frames, METH_VARARGS,
     "decode frames -> [(seq, tag, body)] with torn-tail truncation"},
    {"crc32", crc32_py, METH_VARARGS, "zlib-compatible slice-by-8 CRC32"},
    {nullptr, nullptr, 0, nullptr}};

PyModuleDef moduledef = {PyModuleDef_HEAD_INIT, "_wal_native",
                         "native WAL frame codec", -1, methods};

}  // namespace

PyMODINIT_FUNC PyInit__wal_native(void) {
  init_crc_tables();
  return PyModule_Create(&moduledef);
}
