import sys, time, multiprocessing as mp
from pathlib import Path
sys.path.insert(0, str(Path(__file__).resolve().parent.parent))


def run_worker(idx, seconds, counter):
    import torch
    torch.set_num_threads(1)
    import threading
    from pdrl_amd.agents import Worker
    from pdrl_amd.transport import Endpoint
    from pdrl_amd.utils import load_params
    import main as main_mod
    p = load_params(); p.algo = "IMPALA"; p.env = "CartPole-v1"
    p.num_envs_per_worker = 4
    main_mod.probe_env_spaces(p)
    model = main_mod.build_model(p)
    sub = Endpoint(bind=("127.0.0.1", 0))
    w = Worker(model, idx, "127.0.0.1", sub.bound_port, "127.0.0.1", 1, p,
               seed=idx)
    stop = threading.Event()
    w.stop_event = stop
    threading.Thread(target=lambda: (time.sleep(seconds), stop.set()),
                     daemon=True).start()
    w.collect()
    counter.value = w._total_steps


if __name__ == "__main__":
    mp.set_start_method("spawn", force=True)
    N = int(sys.argv[1]) if len(sys.argv) > 1 else 16
    secs = 8
    cs = [mp.Value("l", 0) for _ in range(N)]
    ps = [mp.Process(target=run_worker, args=(i, secs, cs[i]))
          for i in range(N)]
    for p_ in ps:
        p_.start()
    for p_ in ps:
        p_.join(timeout=90)
    tot = sum(c.value for c in cs)
    print(f"{N} workers x M=4: {tot} steps in {secs}s -> "
          f"{tot/secs:.0f} steps/s aggregate ({tot/secs/N:.0f}/worker)")
