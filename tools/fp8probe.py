import ctypes, numpy as np, torch, sys
sys.path.insert(0, "/root/repo")
import flashmoe_amd._ext as _ext
lib = _ext.load()
inp = torch.arange(256, dtype=torch.uint8, device="cuda")
out = torch.zeros(256, dtype=torch.float32, device="cuda")
r = lib.fm_debug_fp8cvt(None, ctypes.c_void_p(inp.data_ptr()), ctypes.c_void_p(out.data_ptr()))
torch.cuda.synchronize()
got = out.cpu().numpy()
want = inp.cpu().view(torch.float8_e4m3fn).float().numpy()
bad = [(i, float(got[i]), float(want[i])) for i in range(256)
       if not (np.isnan(got[i]) and np.isnan(want[i])) and got[i] != want[i]]
print("rc", r, "mismatches:", len(bad))
for b in bad[:12]: print(b)
