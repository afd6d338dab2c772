import ctypes as C
import sys
import torch
sys.path.insert(0, "/root/repo")
from opentenbase_amd import executor as ex
from opentenbase_amd._lib import call, lib
ex.init_device(0)
n, ngroups = 600_000_000, 100_000_000
g = torch.Generator(device="cuda").manual_seed(1)
keys = torch.randint(0, ngroups, (n,), dtype=torch.int64, device="cuda", generator=g)
vals = torch.rand(n, dtype=torch.float64, device="cuda", generator=g)
L = lib()
ws_bytes = C.c_size_t(0)
L.otbx_agg_i64_workspace_bytes(C.c_int64(n), C.byref(ws_bytes))
ws = torch.empty(ws_bytes.value, dtype=torch.uint8, device="cuda")
out = torch.empty(n * 40, dtype=torch.uint8, device="cuda")
ng = torch.zeros(1, dtype=torch.int64, device="cuda")
stream = C.c_void_p(torch.cuda.current_stream().cuda_stream)
torch.cuda.synchronize()
for _ in range(3):
    call("otbx_agg_i64", C.c_void_p(keys.data_ptr()), None,
         C.c_void_p(vals.data_ptr()), None, C.c_int64(n),
         C.c_void_p(ws.data_ptr()), C.c_size_t(ws_bytes.value),
         C.c_void_p(out.data_ptr()), C.c_void_p(ng.data_ptr()), stream)
    torch.cuda.synchronize()
print("ngroups", int(ng.cpu().item()))
