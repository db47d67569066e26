"""Cross-rank merge of LeafSearchResponse protobufs.

This is the root-side reduce of the reference (merge_fruits,
quickwit-search/src/collector.rs:832-861 + intermediate aggregation merge)
applied across GPUs: each rank leaf-searches its split batch, the packed
responses are exchanged (RCCL/gloo via torch.distributed in bench.py), and
rank 0 merges with identical tie-break semantics through the product C-ABI
(qw_merge_leaf_responses — ctx-less, CPU-side, no GPU needed)."""
import ctypes
import os

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
_lib = None


class _Buf(ctypes.Structure):
    _fields_ = [("data", ctypes.POINTER(ctypes.c_uint8)), ("len", ctypes.c_size_t)]


def _get_lib():
    global _lib
    if _lib is None:
        _lib = ctypes.CDLL(os.path.join(REPO_ROOT, "libquickwit_amd.so"))
        _lib.qw_merge_leaf_responses.argtypes = [
            ctypes.c_char_p, ctypes.c_size_t,
            ctypes.POINTER(ctypes.c_char_p), ctypes.POINTER(ctypes.c_size_t),
            ctypes.c_size_t, ctypes.POINTER(_Buf)]
        _lib.qw_buf_free.argtypes = [ctypes.POINTER(_Buf)]
    return _lib


def merge_leaf_responses(search_request_pb: bytes, response_pbs: list) -> bytes:
    lib = _get_lib()
    arr = (ctypes.c_char_p * len(response_pbs))(*response_pbs)
    lens = (ctypes.c_size_t * len(response_pbs))(*[len(r) for r in response_pbs])
    buf = _Buf()
    rc = lib.qw_merge_leaf_responses(search_request_pb, len(search_request_pb),
                                     arr, lens, len(response_pbs), ctypes.byref(buf))
    if rc != 0:
        raise RuntimeError(f"qw_merge_leaf_responses failed ({rc})")
    out = ctypes.string_at(buf.data, buf.len)
    lib.qw_buf_free(ctypes.byref(buf))
    return out
