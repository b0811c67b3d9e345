"""Probe hipIpcGetMemHandle across sizes + mesh realloc scenario."""
import sys
sys.path.insert(0, __file__.rsplit('/', 2)[0])
import ctypes
import torch  # loads the hip runtime
lib = ctypes.CDLL("libamdhip64.so.7" if False else None, use_errno=True)
import gloo_amd as ga

hip = ctypes.CDLL("/usr/local/lib/python3.10/dist-packages/torch/lib/libamdhip64.so")
hip.hipMalloc.argtypes = [ctypes.POINTER(ctypes.c_void_p), ctypes.c_size_t]
handle = (ctypes.c_char * 64)()

def probe(sz):
    p = ctypes.c_void_p()
    rc = hip.hipMalloc(ctypes.byref(p), sz)
    rc2 = hip.hipIpcGetMemHandle(handle, p)
    print(f"size {sz>>20}MB: malloc rc={rc} ipc rc={rc2}", flush=True)
    hip.hipFree(p)

torch.cuda.init()
for sz in [1<<22, 1<<25, 100<<20, 208<<20, 210<<20, 1<<30]:
    probe(sz)
# alloc-free-alloc pattern
a = ctypes.c_void_p(); hip.hipMalloc(ctypes.byref(a), 8<<20)
print("first:", hip.hipIpcGetMemHandle(handle, a))
hip.hipFree(a)
b = ctypes.c_void_p(); hip.hipMalloc(ctypes.byref(b), 208<<20)
print("realloc:", hip.hipIpcGetMemHandle(handle, b))
