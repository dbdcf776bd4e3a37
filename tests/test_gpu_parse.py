"""GPU numerics tests: HIP parse+encode kernels vs the pure-Python oracle.

Every test compares device output byte-for-byte / set-for-set against
crawler_amd.ops.golden_batch (reference semantics of
telegramhelper/tdutils.go — see golden.py citations).
"""
import datetime as dt

import numpy as np
import pytest
import torch

from crawler_amd.feed import FeedConfig, SyntheticFeed
from crawler_amd.ops import batch as B
from crawler_amd.ops import golden as G
from crawler_amd.ops.golden_batch import encode_batch

pytestmark = pytest.mark.gpu

UTC = dt.timezone.utc
NOW = dt.datetime(2026, 2, 3, 4, 5, 6, tzinfo=UTC)


@pytest.fixture(scope="module")
def gpu_mod():
    from crawler_amd.ops import gpu

    gpu.require_lib()
    return gpu


@pytest.fixture(scope="module")
def feed():
    return SyntheticFeed(FeedConfig(seed=99, universe=50_000))


def _roundtrip(gpu_mod, batch, now=NOW, min_post_date=None):
    golden_lines, golden_links = encode_batch(
        batch, now=now, min_post_date=min_post_date
    )
    res = gpu_mod.parse_encode(
        batch.to("cuda:0"), now=now, min_post_date=min_post_date
    )
    torch.cuda.synchronize()
    return golden_lines, golden_links, res


def test_bytes_identical_small(gpu_mod, feed):
    batch = feed.build_batch(np.arange(8), posts_per_channel=128)
    golden_lines, _, res = _roundtrip(gpu_mod, batch)
    out = bytes(res.out.cpu().numpy())
    expect = b"".join(golden_lines)
    if out != expect:
        # pinpoint the first divergent line for debuggability
        off = 0
        lens = res.line_len.cpu().numpy()
        offs = res.line_off.cpu().numpy()
        for i, gl in enumerate(golden_lines):
            dev_line = out[offs[i]: offs[i] + lens[i]]
            assert dev_line == gl, (
                f"line {i} differs:\nGPU: {dev_line[:400]!r}\n"
                f"CPU: {gl[:400]!r}"
            )
    assert out == expect


def test_line_lengths_match(gpu_mod, feed):
    batch = feed.build_batch(np.arange(4), posts_per_channel=256)
    golden_lines, _, res = _roundtrip(gpu_mod, batch)
    lens = res.line_len.cpu().numpy()
    for i, gl in enumerate(golden_lines):
        assert lens[i] == len(gl), f"line {i}: {lens[i]} != {len(gl)}"


def test_links_match_golden(gpu_mod, feed):
    batch = feed.build_batch(np.arange(16), posts_per_channel=64)
    _, golden_links, res = _roundtrip(gpu_mod, batch)
    dev_links = gpu_mod.links_to_python(res)
    for i in range(batch.n):
        assert dev_links[i] == golden_links[i], (
            f"msg {i}: GPU {dev_links[i]} != CPU {golden_links[i]}"
        )


def test_min_post_date_filter(gpu_mod, feed):
    batch = feed.build_batch(np.arange(2), posts_per_channel=32)
    cutoff = dt.datetime.fromtimestamp(
        int(batch.meta["date"][40]), UTC
    )
    golden_lines, _, res = _roundtrip(gpu_mod, batch, min_post_date=cutoff)
    lens = res.line_len.cpu().numpy()
    n_zero = int((lens == 0).sum())
    n_golden_zero = sum(1 for l in golden_lines if l == b"")
    assert n_zero == n_golden_zero > 0
    out = bytes(res.out.cpu().numpy())
    assert out == b"".join(golden_lines)


def test_handcrafted_edge_cases(gpu_mod):
    """Pack adversarial messages through the real packer and compare."""
    msgs = []
    texts = [
        # overlap/cursor semantics
        "t.me/abcdet.me/xyz12 tail",
        # 32-char cap
        "t.me/" + "a" + "b" * 40,
        # reserved + valid
        "t.me/joinchat/xx t.me/okchan1",
        # https prefix + dup
        "https://t.me/dupdup1 t.me/dupdup1",
        # unicode + escapes + U+2028
        'привет "мир" <>&\n  t.me/unichan1 🚀',
        # no links
        "just plain text",
        # mention entity with cyrillic prefix
        "канал @mention_chan тут",
    ]
    for k, t in enumerate(texts):
        ents = []
        if "@mention_chan" in t:
            off = t.index("@")
            ents.append(G.Entity("mention", off, len("@mention_chan")))
        msgs.append(G.SynthMessage(
            chat_id=-100123, msg_id=(k + 1) << 20, date=1_700_000_000 + k,
            content_type="messageText",
            text=G.FormattedText(text=t, entities=ents),
            views=k * 10, forwards=k, reactions={"👍": k + 1},
            poster_handle=f"user{k:04d}",
        ))
    ch = [B.ChannelRow(chat_id=-100123, username="edgechan1",
                       title='Edge "Chan" <&>', member_count=5,
                       post_count=len(msgs), total_views=100)]
    batch = B.pack(msgs, ch, [0] * len(msgs))
    from crawler_amd.ops import gpu

    golden_lines, golden_links = encode_batch(batch, now=NOW)
    res = gpu.parse_encode(batch.to("cuda:0"), now=NOW)
    torch.cuda.synchronize()
    out = bytes(res.out.cpu().numpy())
    offs = res.line_off.cpu().numpy()
    lens = res.line_len.cpu().numpy()
    for i, gl in enumerate(golden_lines):
        dev_line = out[offs[i]: offs[i] + lens[i]]
        assert dev_line == gl, (
            f"edge case {i} ({texts[i][:40]!r}):\n"
            f"GPU: {dev_line!r}\nCPU: {gl!r}"
        )
    assert gpu.links_to_python(res) == golden_links


def test_larger_batch_bytes_identical(gpu_mod, feed):
    batch = feed.build_batch(np.arange(40), posts_per_channel=250)
    golden_lines, _, res = _roundtrip(gpu_mod, batch)
    out = bytes(res.out.cpu().numpy())
    assert out == b"".join(golden_lines)


def test_swar_block_boundary_escapes(gpu_mod):
    """Adversarial placement of escape bytes around the measure pass's
    256-byte SWAR blocks (common.h esc fast path): U+2028/29 leaders at
    block edges, specials at every lane-word position, long clean runs."""
    U2028 = " "
    U2029 = " "
    texts = []
    # E2-leader at byte offsets straddling the 256-byte block boundary.
    # description starts at the raw text start, so byte index within the
    # field == byte index within the python string's utf-8 encoding.
    for lead in (253, 254, 255, 256, 257):
        t = "a" * lead + U2028 + "b" * 300
        texts.append(t)
        texts.append("a" * lead + U2029 + "b" * 300)
    # special byte at each position of the first lane word
    for pos in (0, 1, 2, 3, 63, 64, 127, 255, 256, 511):
        t = ("x" * pos) + '"' + ("y" * 400)
        texts.append(t)
    # a control byte deep in an otherwise clean 1KB run
    texts.append("c" * 700 + "\x07" + "d" * 300)
    # fully clean 1KB (pure fast path), clean multiple-of-256
    texts.append("e" * 1024)
    texts.append("f" * 512)
    # dirty first block then long clean tail (fast path must re-engage)
    texts.append("<&>" + "g" * 900)
    # non-ascii high bytes that are NOT E2 sequences (must stay clean-ish)
    texts.append("п" * 400)   # 0xD0 0xBF pairs
    # E2 that is NOT a U+2028/29 (e.g. '…' U+2026 = E2 80 A6)
    texts.append("h" * 250 + "…" + "i" * 300)

    msgs = []
    for k, t in enumerate(texts):
        msgs.append(G.SynthMessage(
            chat_id=-100555, msg_id=(k + 1) << 20,
            date=1_700_100_000 + k, content_type="messageText",
            text=G.FormattedText(text=t),
            views=k, forwards=0, poster_handle=f"user{k:04d}",
        ))
    ch = [B.ChannelRow(chat_id=-100555, username="swarchan1",
                       title="SWAR", member_count=3,
                       post_count=len(msgs), total_views=9)]
    batch = B.pack(msgs, ch, [0] * len(msgs))
    from crawler_amd.ops import gpu

    golden_lines, _ = encode_batch(batch, now=NOW)
    res = gpu.parse_encode(batch.to("cuda:0"), now=NOW)
    torch.cuda.synchronize()
    out = bytes(res.out.cpu().numpy())
    offs = res.line_off.cpu().numpy()
    lens = res.line_len.cpu().numpy()
    for i, gl in enumerate(golden_lines):
        dev_line = out[offs[i]: offs[i] + lens[i]]
        assert dev_line == gl, f"swar case {i}: {texts[i][:50]!r}"


def test_comment_heavy_posts_bytes_identical(gpu_mod):
    """BASELINE config #4 shape (--max-comments 1000): posts carrying
    >100 comments each encode byte-identically to the oracle
    (VERDICT r01 item 6)."""
    heavy = SyntheticFeed(FeedConfig(
        seed=31, universe=5000, comment_rate=1.0,
        max_comments_per_post=250,
    ))
    batch = heavy.build_batch(np.arange(6), posts_per_channel=24)
    # the corpus really exercises the >100-comment shape
    max_c = int(batch.meta["com_cnt"].max())
    assert max_c > 100, f"synth corpus too light: max {max_c} comments"
    golden_lines, _, res = _roundtrip(gpu_mod, batch)
    out = bytes(res.out.cpu().numpy())
    expect = b"".join(golden_lines)
    lens = res.line_len.cpu().numpy()
    offs = res.line_off.cpu().numpy()
    for i, gl in enumerate(golden_lines):
        assert out[offs[i]: offs[i] + lens[i]] == gl, f"line {i} differs"
    assert out == expect


def test_comment_heavy_device_feedgen_matches_host(gpu_mod):
    """Device-side feed generation agrees with the host generator for
    comment-heavy batches (the bench's corpus path)."""
    cfgkw = dict(seed=31, universe=5000, comment_rate=1.0,
                 max_comments_per_post=250)
    heavy = SyntheticFeed(FeedConfig(**cfgkw))
    cids = np.arange(6)
    host = heavy.build_batch(cids, posts_per_channel=24)
    dev = heavy.build_batch_device(cids, torch.device("cuda:0"),
                                   posts_per_channel=24)
    torch.cuda.synchronize()
    gl_host, _ = encode_batch(host, now=NOW)
    res = gpu_mod.parse_encode(dev, now=NOW)
    torch.cuda.synchronize()
    assert bytes(res.out.cpu().numpy()) == b"".join(gl_host)


def _writer_pick_corpus():
    """A batch with one >8KB text (over the staged writer's LDS budget)
    plus normal posts — forces the plain-writer fallback."""
    big = ("Huge " + "x" * 8000 + ' with "escapes" <&> and t.me/bigchan1 '
           + "я" * 120)
    msgs = []
    for k, t in enumerate([big, "small t.me/smallchan1", "plain"]):
        msgs.append(G.SynthMessage(
            chat_id=-100777, msg_id=(k + 1) << 20,
            date=1_700_000_100 + k, content_type="messageText",
            text=G.FormattedText(text=t), views=k, forwards=k,
            reactions={"🔥": k + 2}, poster_handle=f"user{k:04d}",
        ))
    ch = [B.ChannelRow(chat_id=-100777, username="bigtextchan",
                       title="Big", member_count=9, post_count=len(msgs),
                       total_views=50)]
    return B.pack(msgs, ch, [0] * len(msgs))


def test_oversize_text_uses_plain_writer_and_matches(gpu_mod):
    """The host gate must route oversize batches to the plain writer
    (staged LDS budget exceeded) with identical bytes."""
    from crawler_amd.ops.gpu import _stage_fits, require_lib

    batch = _writer_pick_corpus()
    dev_batch = batch.to("cuda:0")
    lib = require_lib()
    # verify the gate actually rejects this batch
    from crawler_amd.ops.gpu import const_tables
    tables = const_tables(dev_batch.device)
    assert not _stage_fits(dev_batch, lib, tables, b"x" * 30, b"y" * 35)
    golden_lines, _ = encode_batch(batch, now=NOW)
    res = gpu_mod.parse_encode(dev_batch, now=NOW)
    torch.cuda.synchronize()
    assert bytes(res.out.cpu().numpy()) == b"".join(golden_lines)


def test_plain_writer_forced_matches_staged(gpu_mod, feed,
                                            monkeypatch):
    """CRAWL_NO_STAGED=1 (plain writer) and the default staged writer
    must produce identical bytes on a staged-eligible corpus."""
    import os

    batch = feed.build_batch(np.arange(4), posts_per_channel=64)
    dev1 = batch.to("cuda:0")
    res_staged = gpu_mod.parse_encode(dev1, now=NOW)
    torch.cuda.synchronize()
    monkeypatch.setenv("CRAWL_NO_STAGED", "1")
    dev2 = batch.to("cuda:0")
    res_plain = gpu_mod.parse_encode(dev2, now=NOW)
    torch.cuda.synchronize()
    monkeypatch.delenv("CRAWL_NO_STAGED")
    assert bytes(res_staged.out.cpu().numpy()) == bytes(
        res_plain.out.cpu().numpy())


def test_thousand_comment_posts_bytes_identical(gpu_mod):
    """The full --max-comments 1000 shape (~57 KB lines): threads up to
    1000 comments, byte-identical to the oracle (the comment tail reads
    global memory — unaffected by the staged writer's LDS budget)."""
    heavy = SyntheticFeed(FeedConfig(
        seed=13, universe=2000, comment_rate=1.0,
        max_comments_per_post=1000,
    ))
    batch = heavy.build_batch(np.arange(3), posts_per_channel=8)
    max_c = int(batch.meta["com_cnt"].max())
    assert max_c > 500, f"corpus too light: max {max_c} comments"
    golden_lines, _, res = _roundtrip(gpu_mod, batch)
    assert bytes(res.out.cpu().numpy()) == b"".join(golden_lines)


def test_single_pass_variant_bytes_identical(gpu_mod, feed):
    """The scratch+compact single-pass path (kept for experimentation)
    stays byte-exact through emitter refactors."""
    batch = feed.build_batch(np.arange(6), posts_per_channel=100)
    golden_lines, _ = encode_batch(batch, now=NOW)
    res = gpu_mod.parse_encode(batch.to("cuda:0"), now=NOW,
                               single_pass=True)
    torch.cuda.synchronize()
    assert bytes(res.out.cpu().numpy()) == b"".join(golden_lines)


def test_entity_search_retries_later_tme(gpu_mod):
    """re.search semantics inside url/text_url entities (fuzz seed 5049
    regression): a 't.me/' with no valid name after it does NOT end the
    search — the next occurrence inside the same entity can still
    match, truncated at the entity boundary ('t.me/joinc' out of
    'joinchat' when the entity ends mid-word). Also pins: first regex
    match DOES end the search even when reserved, and the plaintext
    FindAll walk consumes reserved matches' spans."""
    t1 = "plain e>f ñé test … t.me/    t.me/joinchat/xyz t.me/"
    ents1 = [
        G.Entity("mention", 14, 9),
        G.Entity("url", 30, 13),
        G.Entity("url", 20, 19),   # covers "t.me/    t.me/joinc"
    ]
    # url entity whose FIRST regex match is reserved: search stops
    # there, so 'okname99' later in the entity is NOT found
    t2 = "x t.me/share t.me/okname99 y"
    ents2 = [G.Entity("url", 0, len(t2))]
    # text_url attribute with a dead t.me/ before a live one
    t3 = "click here"
    ents3 = [G.Entity("text_url", 0, 5,
                      url="https://t.me/ %% t.me/attrchan9 z")]
    msgs = []
    for k, (t, ents) in enumerate([(t1, ents1), (t2, ents2),
                                   (t3, ents3)]):
        msgs.append(G.SynthMessage(
            chat_id=-100123, msg_id=(k + 1) << 20,
            date=1_700_000_000 + k, content_type="messageText",
            text=G.FormattedText(text=t, entities=ents),
            views=k, forwards=k, poster_handle=f"user{k:04d}",
        ))
    ch = [B.ChannelRow(chat_id=-100123, username="entchan99",
                       title="T", member_count=5, post_count=len(msgs),
                       total_views=9)]
    batch = B.pack(msgs, ch, [0] * len(msgs))
    from crawler_amd.ops import gpu

    golden_lines, golden_links = encode_batch(batch, now=NOW)
    # oracle sanity: the exact expectations this test exists for
    assert ("joinc", "url") in golden_links[0]
    # the url-entity search stopped at the reserved first match, so
    # okname99 is attributed to the PLAINTEXT scan, not the entity
    assert ("okname99", "plaintext") in golden_links[1]
    assert ("okname99", "url") not in golden_links[1]
    assert ("attrchan9", "text_url") in golden_links[2]
    res = gpu.parse_encode(batch.to("cuda:0"), now=NOW)
    torch.cuda.synchronize()
    out = bytes(res.out.cpu().numpy())
    offs = res.line_off.cpu().numpy()
    lens = res.line_len.cpu().numpy()
    for i, gl in enumerate(golden_lines):
        assert out[offs[i]: offs[i] + lens[i]] == gl, f"msg {i}"
