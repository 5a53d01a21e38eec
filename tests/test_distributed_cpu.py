"""Multi-process distributed tests on CPU (gloo, world_size=2).

Covers the DDP-wrapped training step, the rank-chunked sampler under a
real process group, and gradient equivalence between DDP and a manual
all-reduce — the distributed path must be correct by construction since
8-GPU runs happen only at round end (reference's gloo smoke harness:
src/dataset.py:431-506).
"""

import json
import os

import numpy as np
import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

WORLD = 2


def _init(rank, world, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)


def _run_ddp_step(rank, world, port, tmpdir, q):
    try:
        _init(rank, world, port)
        torch.manual_seed(0)  # same init on all ranks
        from bert_pytorch_amd.config import BertConfig
        from bert_pytorch_amd.models import (
            BertForPreTraining,
            BertPretrainingCriterion,
        )

        config = BertConfig(
            vocab_size_or_config_json_file=256, hidden_size=32,
            num_hidden_layers=2, num_attention_heads=2,
            intermediate_size=64, max_position_embeddings=32,
        )
        model = BertForPreTraining(config)
        ddp = torch.nn.parallel.DistributedDataParallel(model)
        criterion = BertPretrainingCriterion(config.vocab_size)

        g = torch.Generator().manual_seed(100 + rank)  # different data/rank
        ids = torch.randint(0, 256, (2, 16), generator=g)
        mask = torch.ones(2, 16, dtype=torch.long)
        labels = torch.full((2, 16), -1, dtype=torch.long)
        labels[:, 3] = 5
        nsp = torch.zeros(2, dtype=torch.long)

        model.eval()  # kill dropout for determinism
        scores, rel, glabels = ddp(ids, None, mask, masked_lm_labels=labels)
        loss = criterion(scores, rel, glabels, nsp)
        loss.backward()

        # DDP grads must equal the manual average of per-rank grads
        solo = BertForPreTraining(config)
        torch.manual_seed(0)
        solo_full = BertForPreTraining(config)
        solo_full.load_state_dict(model.state_dict())
        solo_full.eval()
        s2, r2, l2 = solo_full(ids, None, mask, masked_lm_labels=labels)
        criterion(s2, r2, l2, nsp).backward()
        name, p_ddp = next(iter(ddp.module.named_parameters()))
        p_solo = dict(solo_full.named_parameters())[name]
        manual = p_solo.grad.clone()
        dist.all_reduce(manual)
        manual /= world
        ok = torch.allclose(p_ddp.grad, manual, rtol=1e-5, atol=1e-6)
        q.put((rank, bool(ok), float(loss)))
        dist.destroy_process_group()
    except Exception as e:  # noqa: BLE001
        q.put((rank, f"ERROR: {e}", 0.0))


@pytest.mark.timeout(120)
def test_ddp_grad_sync_gloo(tmp_path):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29511
    procs = [
        ctx.Process(target=_run_ddp_step, args=(r, WORLD, port, str(tmp_path), q))
        for r in range(WORLD)
    ]
    for p in procs:
        p.start()
    results = [q.get(timeout=110) for _ in range(WORLD)]
    for p in procs:
        p.join(timeout=30)
    for rank, ok, loss in results:
        assert ok is True, f"rank {rank}: {ok}"


def _run_sampler_world(rank, world, port, data_dir, q):
    try:
        _init(rank, world, port)
        from bert_pytorch_amd.data import (
            DistributedSampler,
            ShardedPretrainingDataset,
        )

        files = sorted(
            os.path.join(data_dir, f) for f in os.listdir(data_dir)
        )
        ds = ShardedPretrainingDataset(
            files, mask_token_index=103, max_pred_per_seq=4,
            masked_lm_prob=0.15, vocab_size=500, seed=0,
        )
        sampler = DistributedSampler(
            ds, num_replicas=world, rank=rank, seed=0
        )
        loader = torch.utils.data.DataLoader(
            ds, sampler=sampler, batch_size=4, num_workers=2, drop_last=True
        )
        seen = []
        for epoch in range(2):
            sampler.set_epoch(epoch)
            for batch in loader:
                seen.append(batch[0].shape)
        q.put((rank, len(seen)))
        dist.destroy_process_group()
    except Exception as e:  # noqa: BLE001
        q.put((rank, f"ERROR: {e}"))


@pytest.mark.timeout(120)
def test_sharded_dataset_multirank_workers(tmp_path):
    from bert_pytorch_amd.data import synth

    synth.make_dataset(
        str(tmp_path), num_shards=4, samples_per_shard=16, seq_len=24,
        vocab_size=500, seed=3,
    )
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29513
    procs = [
        ctx.Process(
            target=_run_sampler_world, args=(r, WORLD, port, str(tmp_path), q)
        )
        for r in range(WORLD)
    ]
    for p in procs:
        p.start()
    results = [q.get(timeout=110) for _ in range(WORLD)]
    for p in procs:
        p.join(timeout=30)
    for rank, count in results:
        assert count == 16, f"rank {rank}: {count}"  # 32 samples/rank / 4 * 2 epochs


def _run_compress_hook(rank, world, port, q):
    try:
        _init(rank, world, port + 40)
        torch.manual_seed(0)
        lin = torch.nn.Linear(32, 32)
        from bert_pytorch_amd.parallel import comm as bpa_comm

        ddp = bpa_comm.wrap_ddp(lin, 0, grad_compress="fp16")
        g = torch.Generator().manual_seed(200 + rank)
        x = torch.randn(8, 32, generator=g)
        ddp(x).sum().backward()
        # expected: fp16-rounded average of per-rank grads
        solo = torch.nn.Linear(32, 32)
        solo.load_state_dict(lin.state_dict())
        solo(x).sum().backward()
        manual = solo.weight.grad.clone()
        dist.all_reduce(manual)
        manual /= world
        ok = torch.allclose(
            lin.weight.grad, manual, rtol=2e-3, atol=2e-3
        )
        q.put((rank, bool(ok)))
        dist.destroy_process_group()
    except Exception as e:  # noqa: BLE001
        q.put((rank, f"ERROR: {e}"))


@pytest.mark.timeout(120)
def test_grad_compress_hook_gloo():
    """fp16-compressed all-reduce comm hook matches the fp32 average
    within half-precision tolerance (2 ranks, gloo)."""
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [
        ctx.Process(target=_run_compress_hook, args=(r, WORLD, 29551, q))
        for r in range(WORLD)
    ]
    for p in procs:
        p.start()
    results = [q.get(timeout=110) for _ in range(WORLD)]
    for p in procs:
        p.join(timeout=30)
    for rank, ok in results:
        assert ok is True, f"rank {rank}: {ok}"


def _run_ddp_fused_loss_step(rank, world, port, tmpdir, q):
    """DDP grad-sync equivalence through the compute_mlm_loss=True path
    (the runner/bench path: model returns the scalar MLM loss and the
    criterion's 0-dim branch adds only NSP)."""
    try:
        _init(rank, world, port)
        torch.manual_seed(0)
        from bert_pytorch_amd.config import BertConfig
        from bert_pytorch_amd.models import (
            BertForPreTraining,
            BertPretrainingCriterion,
        )

        config = BertConfig(
            vocab_size_or_config_json_file=256, hidden_size=32,
            num_hidden_layers=2, num_attention_heads=2,
            intermediate_size=64, max_position_embeddings=32,
        )
        model = BertForPreTraining(config)
        ddp = torch.nn.parallel.DistributedDataParallel(model)
        criterion = BertPretrainingCriterion(config.vocab_size)

        g = torch.Generator().manual_seed(100 + rank)
        ids = torch.randint(0, 256, (2, 16), generator=g)
        mask = torch.ones(2, 16, dtype=torch.long)
        labels = torch.full((2, 16), -1, dtype=torch.long)
        labels[:, 3] = 5
        nsp = torch.zeros(2, dtype=torch.long)

        model.eval()
        mlm_loss, rel, glabels = ddp(
            ids, None, mask, masked_lm_labels=labels,
            max_predictions_per_seq=4, compute_mlm_loss=True,
        )
        assert mlm_loss.dim() == 0
        loss = criterion(mlm_loss, rel, glabels, nsp)
        loss.backward()

        torch.manual_seed(0)
        solo = BertForPreTraining(config)
        solo.load_state_dict(model.state_dict())
        solo.eval()
        s2, r2, l2 = solo(
            ids, None, mask, masked_lm_labels=labels,
            max_predictions_per_seq=4, compute_mlm_loss=True,
        )
        criterion(s2, r2, l2, nsp).backward()
        # check the tied embedding/decoder weight AND the MLM bias —
        # both only get grads through the fused-loss path
        checked = 0
        ok = True
        for name, p_ddp in ddp.module.named_parameters():
            if name not in (
                "bert.embeddings.word_embeddings.weight",
                "cls.predictions.bias",
            ):
                continue
            p_solo = dict(solo.named_parameters())[name]
            manual = p_solo.grad.clone()
            dist.all_reduce(manual)
            manual /= world
            ok = ok and torch.allclose(
                p_ddp.grad, manual, rtol=1e-5, atol=1e-6
            )
            checked += 1
        q.put((rank, bool(ok and checked == 2), float(loss)))
        dist.destroy_process_group()
    except Exception as e:  # noqa: BLE001
        q.put((rank, f"ERROR: {e}", 0.0))


@pytest.mark.timeout(120)
def test_ddp_fused_mlm_loss_grad_sync_gloo(tmp_path):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29517
    procs = [
        ctx.Process(
            target=_run_ddp_fused_loss_step,
            args=(r, WORLD, port, str(tmp_path), q),
        )
        for r in range(WORLD)
    ]
    for p in procs:
        p.start()
    results = [q.get(timeout=110) for _ in range(WORLD)]
    for p in procs:
        p.join(timeout=30)
    for rank, ok, loss in results:
        assert ok is True, f"rank {rank}: {ok}"
