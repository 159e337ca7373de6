"""Embeddings (CBOW hsoftmax/negsample), GMM, PLSA tests."""

import pytest
import torch

from lightctr_amd.models.embedding import (EmbedHyper, EmbedModel,
                                           build_huffman, vocab_from_tokens)
from lightctr_amd.models.gmm import GMMHyper, GMMModel
from lightctr_amd.models.plsa import PLSAHyper, PLSAModel


def _toy_corpus(n=20000, seed=0):
    """Two clusters of co-occurring words: a0..a4 and b0..b4."""
    g = torch.Generator().manual_seed(seed)
    toks = []
    for _ in range(n // 10):
        grp = "a" if torch.rand(1, generator=g) < 0.5 else "b"
        perm = torch.randperm(5, generator=g)
        toks.extend(f"{grp}{int(i)}" for i in perm)
        toks.extend(f"{grp}{int(i)}" for i in torch.randperm(5, generator=g))
    return toks


def test_build_huffman_prefix_free():
    codes, paths, n_inner = build_huffman([5, 3, 2, 1, 1])
    assert n_inner == 4
    strs = ["".join(map(str, c)) for c in codes]
    for i, a in enumerate(strs):
        for j, b in enumerate(strs):
            if i != j:
                assert not b.startswith(a)
    # frequent words get shorter codes
    assert len(codes[0]) <= len(codes[4])


@pytest.mark.parametrize("loss", ["negsample", "hsoftmax"])
def test_embedding_learns_cooccurrence(loss):
    toks = _toy_corpus()
    vocab, counts = vocab_from_tokens(toks)
    m = EmbedModel(vocab, counts,
                   EmbedHyper(dim=16, window=3, loss=loss, lr=2.0,
                              subsample_t=1e2))  # no subsampling for test
    ids = torch.tensor([m.word2id[t] for t in toks])
    m.train_stream(ids, epochs=6, batch=1024)
    m.normalize()
    # within-group similarity should beat cross-group
    sims = m.E @ m.E.t()
    a = [m.word2id[f"a{i}"] for i in range(5)]
    b = [m.word2id[f"b{i}"] for i in range(5)]
    within = (sims[a][:, a].sum() - 5) / 20 + (sims[b][:, b].sum() - 5) / 20
    cross = sims[a][:, b].mean() * 2
    assert within / 2 > cross / 2 + 0.1, (float(within / 2), float(cross / 2))


def test_embedding_pq_and_pretrain(tmp_path):
    toks = _toy_corpus(n=2000)
    vocab, counts = vocab_from_tokens(toks)
    m = EmbedModel(vocab, counts, EmbedHyper(dim=16, window=2))
    ids = torch.tensor([m.word2id[t] for t in toks])
    m.train_stream(ids, epochs=1)
    pq, codes = m.quantize(n_sub=4)
    assert codes.shape == (len(vocab), 4)
    p = str(tmp_path / "emb.pt")
    m.save(p)
    m2 = EmbedModel.load_pretrain(p, EmbedHyper(dim=16))
    assert torch.allclose(m2.E, m.E)
    assert m2.most_similar(vocab[0], 2)


def test_gmm_recovers_clusters():
    g = torch.Generator().manual_seed(3)
    X = torch.cat([torch.randn(300, 2, generator=g) * 0.3 + 3,
                   torch.randn(300, 2, generator=g) * 0.3 - 3,
                   torch.randn(300, 2, generator=g) * 0.3
                   + torch.tensor([3.0, -3.0])])
    m = GMMModel(GMMHyper(n_components=3, max_iters=60))
    m.fit(X)
    lbl = m.predict(X)
    # purity: each true cluster maps to a dominant component
    purity = 0
    for i in range(3):
        seg = lbl[i * 300:(i + 1) * 300]
        purity += int(seg.bincount(minlength=3).max())
    assert purity / 900 > 0.95
    assert abs(float(m.weights.sum()) - 1) < 1e-5


def test_plsa_separates_topics():
    g = torch.Generator().manual_seed(5)
    # 2 topics with disjoint vocab halves; 100 docs each
    docs, words, cnts = [], [], []
    for d in range(200):
        topic = d % 2
        for _ in range(30):
            w = int(torch.randint(0, 10, (1,), generator=g)) + topic * 10
            docs.append(d)
            words.append(w)
            cnts.append(1.0)
    m = PLSAModel(PLSAHyper(n_topics=2, max_iters=40))
    m.fit(torch.tensor(docs), torch.tensor(words), torch.tensor(cnts),
          n_docs=200, n_words=20)
    zd = m.doc_topics().argmax(dim=1)
    even = zd[::2].float().mean()
    odd = zd[1::2].float().mean()
    assert abs(float(even) - float(odd)) > 0.9  # docs split by parity
    # topic-word distributions concentrate on their half
    t0 = m.p_w_z[int(zd[0])]
    assert float(t0[:10].sum()) > 0.9 or float(t0[10:].sum()) > 0.9


def test_proc_text_topic_tool(tmp_path):
    """Text prep parity with reference data/proc_text_topic.py: vocab by
    frequency (stopwords/markup dropped), dense count rows, and the
    sparse triples feed PLSAModel.fit directly."""
    import subprocess
    import sys

    src = tmp_path / "corpus.txt"
    src.write_text(
        "<doc id=1>\n"
        "The cat sat on the mat with a cat\n"
        "dogs and cats play ball\n"
        "<doc id=2>\n"
        "ball games are fun fun fun\n"
        "the mat was red\n")
    vocab = tmp_path / "vocab.txt"
    train = tmp_path / "train.csv"
    rc = subprocess.run(
        [sys.executable, "tools/proc_text_topic.py", str(src), "10",
         "--vocab-out", str(vocab), "--train-out", str(train)],
        capture_output=True, text=True)
    assert rc.returncode == 0, rc.stderr
    lines = vocab.read_text().splitlines()
    assert 0 < len(lines) <= 10
    terms = {ln.split()[1]: (int(ln.split()[0]), int(ln.split()[2]))
             for ln in lines}
    assert "the" not in terms and "cat" in terms
    assert terms["fun"][1] == 3  # corpus count survives
    rows = train.read_text().splitlines()
    assert len(rows) == 4  # markup lines dropped
    assert all(len(r.split()) == len(lines) for r in rows)

    # sparse mode triples train PLSA end-to-end
    sp = tmp_path / "sparse.csv"
    rc = subprocess.run(
        [sys.executable, "tools/proc_text_topic.py", str(src), "10",
         "--vocab-out", str(vocab), "--train-out", str(sp), "--sparse"],
        capture_output=True, text=True)
    assert rc.returncode == 0, rc.stderr
    import torch

    from lightctr_amd.models.plsa import PLSAHyper, PLSAModel

    trip = [tuple(map(int, ln.split())) for ln in
            sp.read_text().splitlines()]
    d = torch.tensor([t[0] for t in trip])
    w = torch.tensor([t[1] for t in trip])
    c = torch.tensor([t[2] for t in trip])
    m = PLSAModel(PLSAHyper(n_topics=2, max_iters=5, seed=1))
    m.fit(d, w, c, n_docs=int(d.max()) + 1, n_words=len(lines))
