"""P/D disaggregation KV-transfer tests: extract→inject roundtrip, TCP
connector, and the full flow — prefill on engine A, transfer, decode-only
on engine B reproduces the single-engine greedy output."""
import threading

import pytest
import torch

from kaito_amd.engine import EngineConfig, LLMEngine, SamplingParams
from kaito_amd.engine.kv_transfer import (KVPayload, TCPConnector, extract_kv,
                                          inject_kv)
from kaito_amd.models import get_model_config
from kaito_amd.parallel.state import init_parallel


@pytest.fixture(autouse=True)
def _p():
    init_parallel(1)


def _engine(**kw):
    base = dict(model=get_model_config("tiny-llama-test"), device="cpu",
                max_num_seqs=4, num_gpu_blocks=64, enforce_eager=True,
                max_model_len=96)
    base.update(kw)
    return LLMEngine(EngineConfig(**base))


def test_extract_inject_roundtrip():
    eng_a = _engine(seed=1)
    eng_b = _engine(seed=1)
    n, bs = 37, 16
    bt_a = eng_a.pool.allocate(3)
    # write recognizable KV into A's blocks
    for li, (kc, vc) in enumerate(eng_a.runner.kv_caches):
        kc[torch.tensor(bt_a)] = float(li + 1)
        vc[torch.tensor(bt_a)] = float(-(li + 1))
    layers = extract_kv(eng_a.runner.kv_caches, bt_a, n, bs)
    assert layers[0][0].shape[1] == n
    bt_b = eng_b.pool.allocate(3)
    inject_kv(eng_b.runner.kv_caches, bt_b, layers, bs)
    for li, (kc, vc) in enumerate(eng_b.runner.kv_caches):
        got = kc[torch.tensor(bt_b)].permute(1, 0, 2, 3).reshape(
            kc.shape[1], -1, kc.shape[3])[:, :n]
        assert torch.all(got == float(li + 1))


def test_tcp_connector_roundtrip():
    srv = TCPConnector(server=True)
    out = {}

    def client_send():
        c = TCPConnector(port=srv.port)
        c.send(KVPayload("req-1", [1, 2, 3],
                         [(torch.ones(2, 3, 4), torch.zeros(2, 3, 4))],
                         first_token=42))

    t = threading.Thread(target=client_send)
    t.start()
    p = srv.recv()
    t.join()
    assert p.request_id == "req-1" and p.first_token == 42
    assert p.token_ids == [1, 2, 3]
    assert torch.all(p.layers[0][0] == 1)


def test_prefill_decode_disaggregation_matches_colocated():
    """Prefill on engine A; move KV to engine B; decode-only on B must equal
    the colocated run."""
    prompt = list(range(30, 60))
    sp = SamplingParams(max_tokens=6, ignore_eos=True)
    colo = _engine(seed=7)
    expect = colo.generate([prompt], sp)[0].output_token_ids

    pre = _engine(seed=7)
    dec = _engine(seed=7)
    # prefill-side: run ONLY the prefill step (max_tokens=1 → first token)
    sid = pre.add_request(prompt, SamplingParams(max_tokens=1, ignore_eos=True))
    while pre.has_unfinished():
        pre.step()
    seq = pre.seqs[sid]
    first = seq.output_token_ids[0]
    # blocks were freed on finish... re-run prefill capturing before finish:
    # instead drive a fresh request and extract before completion
    pre2 = _engine(seed=7)
    sid2 = pre2.add_request(prompt, SamplingParams(max_tokens=2,
                                                   ignore_eos=True))
    pre2.step()          # prefill (samples first token)
    seq2 = pre2.seqs[sid2]
    layers = extract_kv(pre2.runner.kv_caches, seq2.block_table,
                        len(prompt), pre2.cfg.block_size)
    payload = KVPayload("r", prompt, layers,
                        first_token=seq2.output_token_ids[0])
    assert payload.first_token == first == expect[0]

    # decode-side: allocate, inject, register a sequence that starts at the
    # first generated token
    from kaito_amd.engine.sequence import Sequence, SeqStatus
    dsid = dec.add_request(prompt, SamplingParams(max_tokens=6,
                                                  ignore_eos=True))
    dseq = dec.seqs[dsid]
    dec.scheduler.waiting.remove(dseq)
    need = dec.pool.blocks_needed(len(prompt) + 2)
    dseq.block_table = dec.pool.allocate(need)
    inject_kv(dec.runner.kv_caches, dseq.block_table, payload.layers,
              dec.cfg.block_size)
    dseq.output_token_ids = [payload.first_token]
    dseq.sched_len = len(prompt) + 1
    dseq.status = SeqStatus.RUNNING
    dec.scheduler.running.append(dseq)
    while dec.has_unfinished():
        dec.step()
    assert dseq.output_token_ids == expect
