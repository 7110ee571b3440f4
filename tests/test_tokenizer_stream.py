"""Tokenizer stream-decoder units: incremental UTF-8 for the byte
tokenizer and prefix-delta for BPE — multi-byte characters surface
exactly once, dangling tails flush."""

from agentainer_amd.engine.tokenizer import ByteTokenizer


def test_byte_stream_decoder_multibyte():
    tok = ByteTokenizer(512)
    text = "héllo ✓ wörld"
    ids = tok.encode(text)
    dec = tok.stream_decoder()
    out = "".join(dec.feed(i) for i in ids) + dec.flush()
    assert out == text


def test_byte_stream_decoder_dangling_tail():
    tok = ByteTokenizer(512)
    ids = tok.encode("ok✓")[:-1]  # cut the 3-byte check mark short
    dec = tok.stream_decoder()
    body = "".join(dec.feed(i) for i in ids)
    assert body == "ok"          # incomplete sequence held back
    tail = dec.flush()
    assert tail == "�"      # flushed as replacement, like decode()


def test_hf_stream_decoder_prefix_delta(tmp_path):
    from tokenizers import Tokenizer, decoders, models, pre_tokenizers, trainers

    from agentainer_amd.engine.tokenizer import HFTokenizer

    t = Tokenizer(models.BPE(unk_token=None))
    t.pre_tokenizer = pre_tokenizers.ByteLevel(add_prefix_space=False)
    t.decoder = decoders.ByteLevel()
    tr = trainers.BpeTrainer(vocab_size=300, special_tokens=["<eos>"],
                             initial_alphabet=pre_tokenizers.ByteLevel.alphabet())
    t.train_from_iterator(["the quick brown fox", "héllo wörld"] * 10, tr)
    p = str(tmp_path / "tokenizer.json")
    t.save(p)
    tok = HFTokenizer(p)
    for text in ("the quick fox", "héllo wörld"):
        ids = tok.encode(text)
        dec = tok.stream_decoder()
        out = "".join(dec.feed(i) for i in ids) + dec.flush()
        assert out == tok.decode(ids)
