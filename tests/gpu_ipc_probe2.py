"""Cross-process re-export probe: does hipIpcGetMemHandle fail on a
fresh allocation after prior IPC open/close activity?"""
import sys, os, subprocess, ctypes, tempfile, struct, time
sys.path.insert(0, __file__.rsplit('/', 2)[0])

def child(role, d):
    import torch
    torch.cuda.init()
    hip = ctypes.CDLL("/usr/local/lib/python3.10/dist-packages/torch/lib/libamdhip64.so")
    handle = (ctypes.c_char * 64)()
    if role == "A":
        a = ctypes.c_void_p(); hip.hipMalloc(ctypes.byref(a), 8<<20)
        rc = hip.hipIpcGetMemHandle(handle, a)
        print("A export1 rc=", rc, flush=True)
        open(d+"/h1","wb").write(bytes(handle))
        while not os.path.exists(d+"/opened"): time.sleep(0.1)
        # peer has our buffer open; export a FRESH allocation
        b = ctypes.c_void_p(); hip.hipMalloc(ctypes.byref(b), 16<<20)
        rc2 = hip.hipIpcGetMemHandle(handle, b)
        print("A export2 (peer has h1 open) rc=", rc2, flush=True)
        open(d+"/h2","wb").write(bytes(handle))
        while not os.path.exists(d+"/closed"): time.sleep(0.1)
        # peer closed both; free first, alloc+export again
        hip.hipFree(a)
        c = ctypes.c_void_p(); hip.hipMalloc(ctypes.byref(c), 24<<20)
        rc3 = hip.hipIpcGetMemHandle(handle, c)
        print("A export3 (after close+free) rc=", rc3, flush=True)
        open(d+"/done","w").write("x")
    else:
        while not os.path.exists(d+"/h1"): time.sleep(0.1)
        h = open(d+"/h1","rb").read()
        hb = (ctypes.c_char * 64).from_buffer_copy(h)
        p = ctypes.c_void_p()
        rc = hip.hipIpcOpenMemHandle(ctypes.byref(p), hb, 2)
        print("B open1 rc=", rc, flush=True)
        open(d+"/opened","w").write("x")
        while not os.path.exists(d+"/h2"): time.sleep(0.1)
        h2 = open(d+"/h2","rb").read()
        hb2 = (ctypes.c_char * 64).from_buffer_copy(h2)
        p2 = ctypes.c_void_p()
        rc2 = hip.hipIpcOpenMemHandle(ctypes.byref(p2), hb2, 2)
        print("B open2 rc=", rc2, flush=True)
        hip.hipIpcCloseMemHandle(p)
        hip.hipIpcCloseMemHandle(p2)
        open(d+"/closed","w").write("x")
        while not os.path.exists(d+"/done"): time.sleep(0.1)

if len(sys.argv) > 1:
    child(sys.argv[1], sys.argv[2])
else:
    d = tempfile.mkdtemp()
    env = dict(os.environ); env["HSA_ENABLE_IPC_MODE_LEGACY"]="0"
    ps = [subprocess.Popen([sys.executable, __file__, role, d], env=env)
          for role in ["A","B"]]
    for p in ps: p.wait(120)
    print("probe done", [p.returncode for p in ps])
