"""Profile the op/collective cost DB on hardware and save it
(mesh_profiling.profile_all -> prof_database.pkl).  Single-GPU profiles the
matmul curve; on a multi-GPU world also the RCCL collectives."""
import sys

sys.path.insert(0, ".")
import alpa_amd as aa
from alpa_amd.mesh_profiling import profile_all


def main():
    aa.init()
    mesh = aa.full_mesh() if aa.world_size() > 1 else None
    db = profile_all(mesh, cluster_key="mi355x")
    out = sys.argv[1] if len(sys.argv) > 1 else "prof_database.pkl"
    if aa.rank() == 0:
        db.save(out)
        r = db.query("mi355x", mesh.shape if mesh else (1, 1))
        c = r.op_curves["matmul_bf16"]
        for s, t in zip(c.sizes, c.times):
            print(f"matmul {s:.2e} flops: {t*1e6:9.1f} us "
                  f"({s/t/1e12:7.1f} TF)")
    aa.shutdown()


if __name__ == "__main__":
    main()
