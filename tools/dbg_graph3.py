import os, sys, torch
sys.path.insert(0, ".")
from torch.utils.data import DataLoader
from pdnlp_amd.config import Args, BertConfig
from pdnlp_amd.data import SyntheticClsDataset
from pdnlp_amd.data.collate import Collate
from pdnlp_amd.engine.trainer import build_training, Trainer
from pdnlp_amd.models import BertForSequenceClassification
from pdnlp_amd.utils import set_seed

case = os.environ.get("CASE", "J")
set_seed(123)
cfg = BertConfig.bert_base_chinese()
cfg.num_hidden_layers = 2
args = Args(); args.amp = True; args.amp_dtype = "bf16"
args.epochs = 1; args.do_dev = False; args.log_every = 50
args.hip_graph = True; args.ckpt_path = "/tmp/m.pt"
ds = SyntheticClsDataset(1024, seq_len=128, learnable=True)
loader = DataLoader(ds, batch_size=32, shuffle=True,
                    collate_fn=Collate(None, 128))
model = BertForSequenceClassification(cfg)
wrapped, opt, scaler, trainer = build_training(args, model=model)

if case == "J":
    # manual: eager steps from dataloader batches, then trainer's capture fn
    it = iter(loader)
    for i in range(3):
        b = next(it)
        out = wrapped(input_ids=b["input_ids"].cuda(),
                      attention_mask=b["attention_mask"].cuda(),
                      token_type_ids=b["token_type_ids"].cuda(),
                      labels=b["label"].cuda())
        out.loss.backward(); opt.step(); opt.zero_grad(set_to_none=True)
    trainer._graph_warm = 3
    b = next(it)
    ok = trainer._maybe_capture_graph(b)
    print("J capture ok:", ok, flush=True)
elif case == "K":
    # trainer.train but monkeypatch _after_step to nothing
    Trainer._after_step = lambda self, *a, **k: None
    trainer.train(loader)
    print("K OK", trainer.global_step, flush=True)
elif case == "L":
    # trainer.train but capture IMMEDIATELY (no eager steps first)
    trainer._graph_warm = 3
    trainer.train(loader)
    print("L OK", trainer.global_step, flush=True)
