"""Model-family CPU rehearsals (tiny configs, ws=2): the config-3/4
training loops run end-to-end through the engine and the loss moves.
Numerics of the full-size models are covered on GPU; this pins the
model code + hook-overlap path on every CPU run."""
import torch

from conftest import run_distributed


def _bert_body(rank, size):
    import axon as hvd
    from axon.models.bert import BertForPreTraining, bert_tiny_config

    torch.manual_seed(0)
    model = BertForPreTraining(bert_tiny_config())
    hvd.broadcast_parameters(model.state_dict(), root_rank=0)
    opt = torch.optim.AdamW(model.parameters(), lr=1e-3)
    opt = hvd.DistributedOptimizer(opt,
                                   named_parameters=model.named_parameters())
    g = torch.Generator().manual_seed(rank)
    ids = torch.randint(0, 128, (2, 16), generator=g)
    labels = ids.clone()
    losses = []
    for _ in range(4):
        opt.zero_grad()
        loss = model.mlm_loss(ids, labels)
        loss.backward()
        opt.step()
        losses.append(loss.item())
    assert losses[-1] < losses[0], losses  # learns the (fixed) batch


def test_bert_tiny_trains_ws2():
    run_distributed(_bert_body, 2)


def _llama_body(rank, size):
    import axon as hvd
    from axon.models.llama import LlamaForCausalLM, llama_tiny_config

    torch.manual_seed(0)
    model = LlamaForCausalLM(llama_tiny_config())
    hvd.broadcast_parameters(model.state_dict(), root_rank=0)
    opt = torch.optim.AdamW(model.parameters(), lr=1e-3)
    opt = hvd.DistributedOptimizer(opt,
                                   named_parameters=model.named_parameters())
    g = torch.Generator().manual_seed(rank)
    ids = torch.randint(0, 256, (2, 24), generator=g)
    losses = []
    for _ in range(4):
        opt.zero_grad()
        loss = model.lm_loss(ids)
        loss.backward()
        opt.step()
        losses.append(loss.item())
    assert losses[-1] < losses[0], losses
    # weights identical across ranks after averaged-grad training
    wsum = torch.tensor([sum(float(p.double().sum())
                             for p in model.parameters())])
    mx = hvd.allreduce(wsum, op=hvd.Max, name="lm.max")
    mn = hvd.allreduce(wsum, op=hvd.Min, name="lm.min")
    assert abs(float(mx - mn)) < 1e-10


def test_llama_tiny_trains_ws2():
    run_distributed(_llama_body, 2)


def _bert_lamb_fp16_body(rank, size):
    """Config-3 algorithmic stack on CPU: BERT(tiny) + FusedLAMB(CPU
    path) + fp16 wire compression through the hook-overlap engine."""
    import axon as hvd
    from axon.models.bert import BertForPreTraining, bert_tiny_config
    from axon.optim import FusedLAMB

    torch.manual_seed(1)
    model = BertForPreTraining(bert_tiny_config())
    hvd.broadcast_parameters(model.state_dict(), root_rank=0)
    opt = FusedLAMB(model.parameters(), lr=2e-3, weight_decay=0.01)
    opt = hvd.DistributedOptimizer(opt,
                                   named_parameters=model.named_parameters(),
                                   compression=hvd.Compression.fp16)
    g = torch.Generator().manual_seed(rank)
    ids = torch.randint(0, 128, (2, 16), generator=g)
    labels = ids.clone()
    losses = []
    for _ in range(4):
        opt.zero_grad()
        loss = model.mlm_loss(ids, labels)
        loss.backward()
        opt.step()
        losses.append(loss.item())
    assert losses[-1] < losses[0], losses
    # lockstep weights despite the fp16 wire
    wsum = torch.tensor([sum(float(p.double().sum())
                             for p in model.parameters())])
    mx = hvd.allreduce(wsum, op=hvd.Max, name="bl.max")
    mn = hvd.allreduce(wsum, op=hvd.Min, name="bl.min")
    assert abs(float(mx - mn)) < 1e-10


def test_bert_lamb_fp16_ws2():
    run_distributed(_bert_lamb_fp16_body, 2)
