import sys
import torch
sys.path.insert(0, ".")
from bench import accuracy_probe
dev = torch.device("cuda")
for sched in ("none", "cosine"):
    for ep in (50, 100):
        acc = accuracy_probe(ep, dev, lr_schedule=sched)
        secs = acc["probe_seconds"]
        print(sched, ep, acc["mean_median_abs_err"], "beats_comp:",
              acc["deepr_beats_comp"], f"{secs}s", flush=True)
