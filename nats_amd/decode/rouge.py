"""ROUGE-N / ROUGE-L scorer — behavioural port of scripts/ROUGE.pl.

Reproduces the Perl scorer's exact arithmetic (ROUGE.pl:71-301):
  * per line pair: clipped n-gram hit counting (ngramScore, ROUGE.pl:234-259)
    or LCS DP (lcs_inner, ROUGE.pl:181-232),
  * recall = hit / model-gram count, precision = hit / peer-gram count,
  * per-line R and P are ROUNDED to 5 decimals (the Perl sprintf "%7.5f")
    BEFORE F = (P*R) / ((1-alpha)*P + alpha*R) with alpha=0.5
    (ROUGE.pl:108-129) — rounding preserved for bit-parity,
  * corpus score = mean of per-line values, printed to 3 decimals.

Divergence note: Perl's split(/\\s+/) yields a leading empty token on lines
with leading whitespace; we use str.split() (awk semantics). Identical on
all well-formed input.
"""

ALPHA = 0.5


def _round5(v):
    return float("%7.5f" % v)


def _ngrams(tokens, n):
    grams = {}
    count = 0
    for i in range(len(tokens) - n + 1):
        g = " ".join(tokens[i:i + n])
        grams[g] = grams.get(g, 0) + 1
        count += 1
    return grams, count


def rouge_n(model_line, peer_line, n, alpha=ALPHA):
    """Per-line ROUGE-N. Returns (R, P, F) rounded to 5 decimals."""
    model_grams, model_cnt = _ngrams(peer_line_tokens(model_line), n)
    peer_grams, peer_cnt = _ngrams(peer_line_tokens(peer_line), n)
    hit = 0
    for g, mc in model_grams.items():
        pc = peer_grams.get(g)
        if pc:
            hit += min(pc, mc)
    r = _round5(hit / model_cnt) if model_cnt else _round5(0)
    p = _round5(hit / peer_cnt) if peer_cnt else _round5(0)
    denom = (1 - alpha) * p + alpha * r
    f = _round5((p * r) / denom) if denom > 0 else _round5(0)
    return r, p, f


def peer_line_tokens(line):
    return line.split()


def _lcs(model_tokens, peer_tokens):
    m, n = len(model_tokens), len(peer_tokens)
    if m == 0:
        return 0, 0, n
    prev = [0] * (n + 1)
    for i in range(1, m + 1):
        cur = [0] * (n + 1)
        mi = model_tokens[i - 1]
        for j in range(1, n + 1):
            if mi == peer_tokens[j - 1]:
                cur[j] = prev[j - 1] + 1
            elif prev[j] >= cur[j - 1]:
                cur[j] = prev[j]
            else:
                cur[j] = cur[j - 1]
        prev = cur
    return prev[n], m, n


def rouge_l(model_line, peer_line, alpha=ALPHA):
    """Per-line ROUGE-L. Returns (R, P, F) rounded to 5 decimals."""
    hit, m_cnt, p_cnt = _lcs(model_line.split(), peer_line.split())
    r = _round5(hit / m_cnt) if m_cnt else _round5(0)
    p = _round5(hit / p_cnt) if p_cnt else _round5(0)
    denom = (1 - alpha) * p + alpha * r
    f = _round5((p * r) / denom) if denom > 0 else _round5(0)
    return r, p, f


def score_files(model_path, peer_path, nsize=1, metric="N", alpha=ALPHA):
    """Corpus-level (avg_R, avg_P, avg_F) like the Perl driver
    (ROUGE.pl:20-56). `metric` is "N" or "L"."""
    rs, ps, fs = [], [], []
    with open(model_path) as fm, open(peer_path) as fp:
        for model_line, peer_line in zip(fm, fp):
            model_line = model_line.rstrip("\n")
            peer_line = peer_line.rstrip("\n")
            if metric == "N":
                r, p, f = rouge_n(model_line, peer_line, nsize, alpha)
            elif metric == "L":
                r, p, f = rouge_l(model_line, peer_line, alpha)
            else:
                raise ValueError("metric must be N or L")
            rs.append(r)
            ps.append(p)
            fs.append(f)
    n = len(rs)
    if n == 0:
        return 0.0, 0.0, 0.0
    return (_round5(sum(rs) / n), _round5(sum(ps) / n), _round5(sum(fs) / n))


def format_report(model_path, peer_path, nsize, metric, alpha=ALPHA):
    """The Perl driver's stdout block (ROUGE.pl:59-69)."""
    r, p, f = score_files(model_path, peer_path, nsize, metric, alpha)
    head = "ROUGE-%s" % (nsize if metric == "N" else "L")
    return "%s\nAve_R | Ave_P | Ave_F\n%.3f\t%.3f\t%.3f\n" % (head, r, p, f)
