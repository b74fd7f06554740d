"""MPT-7B single-GPU capacity check (BASELINE config 5): full model +
AdamW state + seq-4096 activations unsharded in 288 GB HBM."""
import sys, time
from pathlib import Path
sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
import torch
from photon_amd.conf import compose, config_yaml_dir
from photon_amd.data.synthetic import SyntheticTokenDataset
from photon_amd.data.shards import StatefulLoader
from photon_amd.models import build_model
from photon_amd.train import Trainer

cfg = compose(config_yaml_dir(), "base", ["llm_config=mpt-7b"]).to_plain()
llm = cfg["llm_config"]
llm["max_seq_len"] = 4096
llm["model"]["max_seq_len"] = 4096
llm["device_train_microbatch_size"] = 4
llm["global_train_batch_size"] = 4   # one microbatch per step for the check
torch.manual_seed(0)
t0 = time.time()
model = build_model(llm)
print(f"build: {time.time()-t0:.1f}s params={sum(p.numel() for p in model.parameters())/1e9:.2f}B")
ds = SyntheticTokenDataset(4096, vocab_size=int(llm["model"]["vocab_size"]), seed=1)
tr = Trainer(model, llm, train_loader=StatefulLoader(ds, 4), device="cuda")
t0 = time.time()
tr.fit("2ba")
torch.cuda.synchronize()
dt = time.time() - t0
print(f"2 steps: {dt:.1f}s  tokens/s={2*4*4096/dt:,.0f}")
print(f"peak HBM: {torch.cuda.max_memory_allocated()/2**30:.1f} GiB of 288")
