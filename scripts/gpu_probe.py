"""Real-hardware dynamic repartition probe (guarded; run on a fresh box).

Exercises the dynamic-MIG-analog path the reference could never ship:
SPX -> CPX -> (enumerate partitions) -> SPX, via the amdsmi HAL.
"""
import sys, time
sys.path.insert(0, ".")
from k8s_dra_driver_amd.hal.amdsmi import AmdSmiDeviceLib

lib = AmdSmiDeviceLib()
lib.open()
g0 = lib.enumerate()[0]
print(f"before: {g0.compute_partition}/{g0.memory_partition} partitions={len(g0.partitions)} renderD{g0.render_minor}")
try:
    t0 = time.time()
    lib.set_compute_partition(0, "CPX")
    print(f"-> CPX ok in {time.time()-t0:.2f}s")
    g0 = lib.enumerate()[0]
    print(f"after CPX: mode={g0.compute_partition} partitions={len(g0.partitions)} "
          f"minors={[p.render_minor for p in g0.partitions]}")
    kfd = [ (n.node_id, n.render_minor, n.gfx_arch) for n in lib.topology.gpu_nodes() ]
    print(f"kfd nodes: {kfd}")
finally:
    t0 = time.time()
    lib.set_compute_partition(0, "SPX")
    print(f"-> restore SPX ok in {time.time()-t0:.2f}s")
    g0 = lib.enumerate()[0]
    print(f"restored: {g0.compute_partition}/{g0.memory_partition} partitions={len(g0.partitions)}")
lib.close()
