"""logreg_loss_grad kernel probe: W-in-registers vs W-in-LDS at the
config-4 shape (5M x 1024, K=2).  Reports achieved TB/s (X bytes / time)
and numerics parity.  NOTE: the SEA_LOGREG_LDS switch is latched at first
launch (static), so the two arms run in separate processes."""
import json
import os
import subprocess
import sys
import time

import torch


def run_arm(name, envvars):
    env = dict(os.environ)
    env.update(envvars)
    env["SEA_ARM"] = name
    r = subprocess.run([sys.executable, __file__, "arm"], env=env,
                       capture_output=True, text=True, cwd="/root/repo")
    assert r.returncode == 0, r.stderr[-1500:]
    return json.loads(r.stdout.strip().splitlines()[-1])


def arm():
    sys.path.insert(0, __import__("os").path.dirname(__import__("os").path.dirname(__import__("os").path.abspath(__file__))))
    from spark_ensemble_amd.ops import dispatch

    m = dispatch._load_hip()
    n, F, K = 5_000_000, 1024, 2
    g = torch.Generator().manual_seed(1)
    x = torch.randn(n, F, generator=g).to("cuda:0")
    y = torch.randint(0, K, (n,), generator=g, dtype=torch.int32).to("cuda:0")
    w = torch.ones(n, device="cuda:0")
    wmat = (0.01 * torch.randn(F + 1, K, generator=g)).to("cuda:0").contiguous()

    def call():
        payload = torch.zeros(1 + (F + 1) * K, dtype=torch.float32,
                              device="cuda:0")
        m.logreg_loss_grad(payload, x, y, w, wmat, True)
        return payload

    call(); torch.cuda.synchronize()
    t0 = time.time()
    reps = 10
    for _ in range(reps):
        p = call()
    torch.cuda.synchronize()
    dt = (time.time() - t0) / reps
    bytes_x = n * F * 4
    print(json.dumps({
        "mode": os.environ.get("SEA_ARM", "?"),
        "ms": round(dt * 1000, 2),
        "tb_per_s": round(bytes_x / dt / 1e12, 2),
        "loss": float(p[0]),
        "grad_head": [float(v) for v in p[1:4]],
    }))


if __name__ == "__main__":
    if len(sys.argv) > 1 and sys.argv[1] == "arm":
        arm()
    else:
        arms = {
            "rr2_lds": {},
            "rr4_lds": {"SEA_LOGREG_RR4": "1"},
            "rr2_wreg": {"SEA_LOGREG_WREG": "1"},
        }
        out = {k: run_arm(k, v) for k, v in arms.items()}
        ref = out["rr2_lds"]["loss"]
        out["loss_match"] = all(
            abs(v["loss"] - ref) / max(abs(ref), 1e-9) < 1e-5
            for k, v in out.items() if isinstance(v, dict)
        )
        print(json.dumps(out))
