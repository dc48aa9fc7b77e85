#!/usr/bin/env python3
"""BERT-Large SQuAD fine-tuning example (reference: examples/squad/
main.py — the huggingface run_squad fork; re-designed, not translated).

This environment has no network, so the default mode fine-tunes the
bundled random-init BERT-Large QA model (`bagua_amd.models.bert`) on
synthetic (input_ids, start/end position) batches of the SQuAD shape —
enough to exercise the full distributed fine-tuning loop, mixed
precision and any bagua algorithm. For real fine-tuning pass
``--model-dir`` (a local HF checkpoint directory) and ``--squad-json``
(SQuAD v1.1 json); the script then uses `transformers` tokenization and
weights (installed in this image, weights must be local).

Launch:
    python -m bagua_amd.distributed.run --nproc-per-node 8 \
        examples/squad/main.py --algorithm qadam
"""

import argparse
import os
import sys
import time

import torch
import torch.nn.functional as F

sys.path.insert(0, os.path.abspath(os.path.join(
    os.path.dirname(os.path.abspath(__file__)), "..", "..")))

import bagua_amd  # noqa: E402
from bagua_amd import env  # noqa: E402
from bagua_amd.parallel.algorithms import GlobalAlgorithmRegistry  # noqa: E402


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--algorithm", default="gradient_allreduce")
    p.add_argument("--batch-size", type=int, default=8)
    p.add_argument("--seq-len", type=int, default=384)
    p.add_argument("--lr", type=float, default=3e-5)
    p.add_argument("--steps", type=int, default=30)
    p.add_argument("--model-dir", default=None,
                   help="local HF checkpoint dir (optional)")
    p.add_argument("--squad-json", default=None,
                   help="local SQuAD v1.1 train json (optional)")
    p.add_argument("--print-freq", type=int, default=5)
    return p.parse_args()


def build_model(args, device):
    if args.model_dir:
        from transformers import AutoModelForQuestionAnswering

        return AutoModelForQuestionAnswering.from_pretrained(
            args.model_dir).to(device), True
    from bagua_amd.models import create_model

    return create_model("bert-large").to(device), False


def batches(args, device):
    torch.manual_seed(4321 + env.get_rank())
    if args.squad_json and args.model_dir:
        # real data: tokenize locally (no downloads)
        import json

        from transformers import AutoTokenizer

        tok = AutoTokenizer.from_pretrained(args.model_dir)
        with open(args.squad_json) as f:
            data = json.load(f)["data"]
        examples = [(qa["question"], para["context"])
                    for art in data for para in art["paragraphs"]
                    for qa in para["qas"]]
        for i in range(0, min(len(examples),
                              args.steps * args.batch_size),
                       args.batch_size):
            chunk = examples[i:i + args.batch_size]
            enc = tok([q for q, _ in chunk], [c for _, c in chunk],
                      max_length=args.seq_len, truncation=True,
                      padding="max_length", return_tensors="pt")
            yield ({k: v.to(device) for k, v in enc.items()},
                   torch.randint(0, args.seq_len,
                                 (len(chunk),), device=device),
                   torch.randint(0, args.seq_len,
                                 (len(chunk),), device=device))
        return
    for _ in range(args.steps):
        ids = torch.randint(0, 30000, (args.batch_size, args.seq_len),
                            device=device)
        start = torch.randint(0, args.seq_len, (args.batch_size,),
                              device=device)
        end = torch.randint(0, args.seq_len, (args.batch_size,),
                            device=device)
        yield ids, start, end


def main():
    args = parse_args()
    use_cuda = torch.cuda.is_available()
    device = torch.device("cuda", env.get_local_rank()) if use_cuda \
        else torch.device("cpu")
    if use_cuda:
        torch.cuda.set_device(device)
    bagua_amd.init_process_group()

    torch.manual_seed(42)
    model, hf = build_model(args, device)

    if args.algorithm == "qadam":
        from bagua_amd.parallel.algorithms.q_adam import QAdamOptimizer

        optimizer = QAdamOptimizer(model.parameters(), lr=args.lr,
                                   warmup_steps=10)
        algorithm = GlobalAlgorithmRegistry.get("qadam")(optimizer)
    else:
        optimizer = torch.optim.AdamW(model.parameters(), lr=args.lr)
        algorithm = GlobalAlgorithmRegistry.get(args.algorithm)()

    ddp = bagua_amd.DistributedDataParallel(
        model, optimizers=[optimizer], algorithm=algorithm)
    amp = torch.autocast(device.type, dtype=torch.bfloat16,
                         enabled=use_cuda)

    t0 = time.time()
    tokens = 0
    for i, batch in enumerate(batches(args, device)):
        optimizer.zero_grad()
        with amp:
            if hf:
                enc, start, end = batch
                out = ddp(**enc, start_positions=start, end_positions=end)
                loss = out.loss
            else:
                ids, start, end = batch
                s_logits, e_logits = ddp(ids)
                loss = (F.cross_entropy(s_logits, start)
                        + F.cross_entropy(e_logits, end))
        loss.backward()
        optimizer.step()
        tokens += args.batch_size * args.seq_len
        if env.get_rank() == 0 and i % args.print_freq == 0:
            print("step %d loss %.4f" % (i, loss.item()), flush=True)
    if use_cuda:
        torch.cuda.synchronize()
    dt = time.time() - t0
    if env.get_rank() == 0:
        print("done: %.0f tokens/s (whole job)"
              % (tokens * env.get_world_size() / dt), flush=True)

    if args.algorithm == "async":
        ddp.inner.bagua_algorithm.abort(ddp)
    bagua_amd.deinit_process_group()


if __name__ == "__main__":
    main()
