"""Direct-Popen murder-fuzz harness (the launcher the murder campaigns
require: torchrun SIGTERMs survivors when any rank dies, Popen lets
them finish).  Kills rank 1 after its 5th op and asserts every
survivor completes and exits 0 while the victim exits 66.

Usage: python tests/murder_popen_harness.py [world] [base_seed ...]
"""
import os
import subprocess
import sys


def run_campaign(world: int, base_seed: int, port: int) -> bool:
    # victim is rank 1, not the LAST rank: data-affinity placement can
    # leave the tail rank under the kill threshold at larger worlds
    # (observed at world 5: rank 4 ran <5 ops in 10 rounds and the
    # injection never fired), while rank 1 always draws work
    victim = 1
    procs = []
    for rank in range(world):
        env = dict(
            os.environ,
            MASTER_ADDR="127.0.0.1",
            MASTER_PORT=str(port),
            RANK=str(rank),
            LOCAL_RANK=str(rank),
            WORLD_SIZE=str(world),
            FUZZ_MURDER=f"{victim}:5",
            FUZZ_ROUNDS="10",
            FUZZ_BASE_SEED=str(base_seed),
        )
        procs.append(
            subprocess.Popen(
                [sys.executable, os.path.join(os.path.dirname(__file__),
                                              "pool_script_fuzz.py")],
                env=env, stdout=subprocess.PIPE,
                stderr=subprocess.STDOUT, text=True,
            )
        )
    outs = [p.communicate()[0] for p in procs]
    rcs = [p.returncode for p in procs]
    ok = rcs[victim] == 66 and all(
        rc == 0 for i, rc in enumerate(rcs) if i != victim
    )
    print(f"base {base_seed}: rcs={rcs} {'OK' if ok else 'FAIL'}", flush=True)
    if not ok:
        for i, o in enumerate(outs):
            print(f"--- rank {i} tail ---\n{o[-2000:]}", flush=True)
    return ok


def main() -> None:
    world = int(sys.argv[1]) if len(sys.argv) > 1 else 3
    seeds = [int(a) for a in sys.argv[2:]] or [98000, 99000]
    for i, seed in enumerate(seeds):
        if not run_campaign(world, seed, 29811 + 2 * i):
            sys.exit(1)
    print("MURDER CAMPAIGN OK", flush=True)


if __name__ == "__main__":
    main()
