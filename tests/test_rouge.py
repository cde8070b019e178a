from nats_amd.decode.replace_unk import replace_unk_line
from nats_amd.decode.rouge import rouge_l, rouge_n, score_files, format_report


def test_rouge1_identical():
    r, p, f = rouge_n("a b c", "a b c", 1)
    assert r == p == f == 1.0


def test_rouge1_known_values():
    # ref: "the cat sat", sys: "the cat" -> hit=2, R=2/3, P=2/2
    r, p, f = rouge_n("the cat sat", "the cat", 1)
    assert abs(r - round(2 / 3, 5)) < 1e-9
    assert p == 1.0
    expect_f = (p * r) / (0.5 * p + 0.5 * r)
    assert abs(f - round(expect_f, 5)) < 1e-4


def test_rouge2_clipping():
    # repeated bigram in sys clipped to ref count
    r, p, f = rouge_n("a b a b", "a b a b a b", 2)
    # ref bigrams: {a b:2, b a:1} cnt=3; sys: {a b:3, b a:2} cnt=5
    # hit = min(2,3)+min(1,2) = 3
    assert abs(r - 1.0) < 1e-9
    assert abs(p - round(3 / 5, 5)) < 1e-9


def test_rouge_l():
    # ref: a b c d ; sys: a x c -> LCS = a c = 2
    r, p, f = rouge_l("a b c d", "a x c")
    assert abs(r - 0.5) < 1e-9
    assert abs(p - round(2 / 3, 5)) < 1e-9


def test_rouge_empty_peer():
    r, p, f = rouge_n("a b", "", 1)
    assert r == 0.0 and p == 0.0 and f == 0.0


def test_score_files_and_report(tmp_path):
    ref = tmp_path / "ref.txt"
    sys_ = tmp_path / "sys.txt"
    ref.write_text("a b c\nx y\n")
    sys_.write_text("a b c\nx z\n")
    r, p, f = score_files(str(ref), str(sys_), 1, "N")
    assert abs(r - round((1.0 + 0.5) / 2, 5)) < 1e-9
    rep = format_report(str(ref), str(sys_), 1, "N")
    assert rep.startswith("ROUGE-1\n")
    assert "Ave_R | Ave_P | Ave_F" in rep


def test_replace_unk_line():
    src = "alpha beta gamma <EOS>".split()
    line = "UNK [1] keep [0] <EOS> [3] UNK [3]"
    # UNK@1 -> beta ; keep stays ; <EOS> dropped ; UNK@3 -> src <EOS> skipped
    out = replace_unk_line(line, src)
    assert out == "beta keep"
