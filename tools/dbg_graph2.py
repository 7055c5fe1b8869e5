import os, sys, torch
sys.path.insert(0, ".")
from torch.utils.data import DataLoader
from pdnlp_amd.config import Args, BertConfig
from pdnlp_amd.data import SyntheticClsDataset
from pdnlp_amd.data.collate import Collate
from pdnlp_amd.engine.trainer import build_training
from pdnlp_amd.models import BertForSequenceClassification
from pdnlp_amd.utils import set_seed

case = os.environ.get("CASE", "F")
set_seed(123)
cfg = BertConfig.bert_base_chinese()
cfg.num_hidden_layers = 2
args = Args()
args.epochs = 1
args.do_dev = False
args.log_every = 50
args.amp = True
args.amp_dtype = "bf16"
args.hip_graph = True
args.learning_rate = 3e-4
args.ckpt_path = "/tmp/m.pt"
if case == "G":
    args.max_grad_norm = 0.0
if case == "H":
    args.log_every = 10**9
ds = SyntheticClsDataset(1024, seq_len=128, learnable=True)
loader = DataLoader(ds, batch_size=32, shuffle=True,
                    collate_fn=Collate(None, 128))
if case == "I":
    import pdnlp_amd.utils.metrics as M
    class _NoT:
        def __init__(self, *a): pass
        def __enter__(self): return self
        def __exit__(self, *a): return False
    M.TraceRange = _NoT
    import pdnlp_amd.engine.trainer as T
    T.TraceRange = _NoT
model = BertForSequenceClassification(cfg)
wrapped, opt, scaler, trainer = build_training(args, model=model)
trainer.train(loader)
print(case, "OK steps", trainer.global_step, "graph", trainer._graph is not None, flush=True)
