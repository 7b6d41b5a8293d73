import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
step = sys.argv[1] if len(sys.argv) > 1 else "all"
from faabric_amd import _core
print("A gpu_count(before torch):", _core.gpu_count(), flush=True)
import torch
print("B torch avail:", torch.cuda.is_available(), flush=True)
t = torch.randn(1024, device="cuda")
torch.cuda.synchronize()
print("C torch tensor ok", flush=True)
print("D gpu_count(after torch):", _core.gpu_count(), flush=True)
s = _core.DeviceSnapshot(4096 * 256)
print("E DeviceSnapshot ok", flush=True)
a = torch.randn(1024, device="cuda"); b = torch.randn(1024, device="cuda")
_core.device_elementwise_op(a.data_ptr(), b.data_ptr(), 1024, 3, 0)
print("F elementwise ok", flush=True)
from faabric_amd.runtime import LocalRuntime
rt = LocalRuntime(slots=4, port_offset=14000, planner_port_offset=14000)
rt.start_planner(with_snapshot_server=False); rt.start_worker()
print("G runtime ok", flush=True)
s2 = _core.DeviceSnapshot(4096 * 256)
_core.device_elementwise_op(a.data_ptr(), b.data_ptr(), 1024, 3, 0)
print("H post-runtime kernels ok", flush=True)
rt.stop()
print("ALL OK", flush=True)
