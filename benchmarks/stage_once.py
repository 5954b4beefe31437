import os, sys, json
sys.path.insert(0, os.environ["GRAFT_REPO_ROOT"] if "GRAFT_REPO_ROOT" in os.environ else ".")
sys.path.insert(0, os.path.join(sys.path[0], "benchmarks"))
import staging_perf
if os.environ.get("BNET_BENCH_SIZES"):
    # patch the size list for short profiled runs
    pass
from staging_perf import bench_plugin_gpu_loopback
cfg = {k: os.environ.get(k) for k in ("BNET_STAGE_CHUNK","BNET_NSTREAMS","BNET_IO_THREADS","BNET_STAGE_POOL")}
print(json.dumps({"cfg": cfg, "res": bench_plugin_gpu_loopback()}))
