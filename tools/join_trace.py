import ctypes as C
import sys
import torch
sys.path.insert(0, "/root/repo")
from opentenbase_amd import executor as ex
from opentenbase_amd._lib import call, lib
ex.init_device(0)
nb, np_ = 150_000_000, 600_000_000
g = torch.Generator(device="cuda").manual_seed(2)
bk = torch.randint(0, nb, (nb,), dtype=torch.int64, device="cuda", generator=g)
pk = torch.randint(0, nb, (np_,), dtype=torch.int64, device="cuda", generator=g)
L = lib()
ws_bytes = C.c_size_t(0)
L.otbx_join_i64_workspace_bytes(C.c_int64(nb), C.c_int64(np_), C.byref(ws_bytes))
ws = torch.empty(ws_bytes.value, dtype=torch.uint8, device="cuda")
cap = int(np_ * 2.2)
ob = torch.empty(cap, dtype=torch.int64, device="cuda")
op = torch.empty(cap, dtype=torch.int64, device="cuda")
npairs = torch.zeros(1, dtype=torch.int64, device="cuda")
stream = C.c_void_p(torch.cuda.current_stream().cuda_stream)
torch.cuda.synchronize()
for _ in range(3):
    call("otbx_join_i64", C.c_void_p(bk.data_ptr()), None, C.c_int64(nb),
         C.c_void_p(pk.data_ptr()), None, C.c_int64(np_),
         C.c_void_p(ws.data_ptr()), C.c_size_t(ws_bytes.value),
         C.c_void_p(ob.data_ptr()), C.c_void_p(op.data_ptr()),
         C.c_int64(cap), C.c_void_p(npairs.data_ptr()), stream)
    torch.cuda.synchronize()
print("pairs", int(npairs.cpu().item()))
