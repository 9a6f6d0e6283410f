import time, torch, os, sys
sys.path.insert(0, "/root/repo")
from pipegoose_amd.ops import get_extension
ext = get_extension(required=True)
torch.manual_seed(0)
x = torch.randn(16384, 4096, device="cuda", dtype=torch.bfloat16)
dy = torch.randn_like(x)
b = torch.randn(4096, device="cuda", dtype=torch.bfloat16)
for _ in range(5): ext.bias_gelu_bwd(dy, x, b)
torch.cuda.synchronize(); t0 = time.perf_counter()
for _ in range(50): ext.bias_gelu_bwd(dy, x, b)
torch.cuda.synchronize()
us = (time.perf_counter()-t0)/50*1e6
gb = 3*x.numel()*2/1e9
print(f"rows={os.environ.get('PG_BGELU_ROWS','512')}: {us:.1f} us  {gb/us*1e6/1000:.2f} TB/s")
