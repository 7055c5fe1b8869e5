import sys, torch
sys.path.insert(0, ".")
from pdnlp_amd.config import Args, BertConfig
from pdnlp_amd.models import BertForSequenceClassification
from pdnlp_amd.ops.adamw import build_optimizer
from pdnlp_amd.utils import set_seed

def try_capture(tag, layers=2, eager_steps=0, save_logits=True, use_bt=False):
    try:
        set_seed(123)
        cfg = BertConfig.bert_base_chinese()
        cfg.num_hidden_layers = layers
        model = BertForSequenceClassification(cfg)
        if use_bt:
            from pdnlp_amd.engine.trainer import build_training
            args = Args(); args.amp = True; args.amp_dtype = "bf16"
            model, opt, _, _tr = build_training(args, model=model)
        else:
            model = model.to(torch.bfloat16).to("cuda")
            opt = build_optimizer(model, lr=3e-4)
        model.train()
        g = torch.Generator().manual_seed(0)
        ids = torch.randint(106, 21128, (32, 128), generator=g).to("cuda")
        mask = torch.ones_like(ids); tids = torch.zeros_like(ids)
        labels = torch.randint(0, 6, (32,), generator=g).to("cuda")
        def fb():
            out = model(input_ids=ids, attention_mask=mask,
                        token_type_ids=tids, labels=labels)
            out.loss.backward()
            return out
        for _ in range(eager_steps):
            fb(); opt.step(); opt.zero_grad(set_to_none=True)
        s = torch.cuda.Stream(); s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(2):
                fb(); opt.step()
        torch.cuda.current_stream().wait_stream(s)
        opt.zero_grad(set_to_none=True)
        graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(graph):
            out = fb()
            keep = (out.loss, out.logits if save_logits else None)
        graph.replay(); opt.step()
        torch.cuda.synchronize()
        print(tag, "OK", flush=True)
    except Exception as e:
        print(tag, "EXC", type(e).__name__, str(e)[:100], flush=True)

import os
which = os.environ.get("CASE")
cases = {
 "A": dict(layers=12, eager_steps=0, save_logits=False, use_bt=False),  # = bench
 "B": dict(layers=2,  eager_steps=0, save_logits=False, use_bt=False),
 "C": dict(layers=2,  eager_steps=3, save_logits=False, use_bt=False),
 "D": dict(layers=2,  eager_steps=3, save_logits=True,  use_bt=False),
 "E": dict(layers=2,  eager_steps=3, save_logits=True,  use_bt=True),
}
try_capture(which, **cases[which])
