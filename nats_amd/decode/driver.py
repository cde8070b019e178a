"""Batch summarization driver — the gen.py engine.

Behavioural port of scripts/gen.py:15-135 with an MI355X-native execution
model: ``n_process`` worker processes each load the full model and decode a
shard of input lines (job/result queues, results re-assembled by index —
gen.py:117-126). On a GPU machine workers round-robin over visible devices
(one process per GPU is the natural MI355X decode layout; the reference
used CPU workers, test.sh:3). Output format is the reference's
"word [srcpos]" interleave (gen.py:88-98), consumed by replace_unk.
"""

import numpy
import torch

from ..data.dictionary import load_dictionary, invert_dictionary
from ..engine.checkpoint import load_checkpoint, load_options
from ..models.distraction import NatsModel
from .beam import gen_sample


def _make_model(model_path, options, device):
    params, _ = load_checkpoint(model_path)
    model = NatsModel(options, params={k: v for k, v in params.items()})
    model.eval()
    return model.to(device)


def _translate_one(model, seq, device, k, normalize, kl_factor, ctx_factor,
                   state_factor):
    x = torch.tensor(seq, dtype=torch.int64, device=device).reshape(-1, 1)
    sample, score, alphas = gen_sample(
        model, x, k=k, maxlen=100, stochastic=False, argmax=False,
        use_unk=True, kl_factor=kl_factor, ctx_factor=ctx_factor,
        state_factor=state_factor)
    score = numpy.array(score)
    if normalize:
        lengths = numpy.array([len(s) for s in sample])
        score = score / lengths
    sidx = int(numpy.argmin(score))
    align_pos = [int(numpy.argmax(alpha)) for alpha in alphas[sidx]]
    return sample[sidx], align_pos


def translate_worker(queue, rqueue, pid, model_path, options, k, normalize,
                     kl_factor, ctx_factor, state_factor, device,
                     maxlen=100):
    """Worker loop (translate_model, gen.py:15-58). Jobs arrive as CHUNKS
    of (idx, ids) pairs and are decoded jointly (batched beams)."""
    from .batched import gen_sample_batched
    model = _make_model(model_path, options, device)
    while True:
        req = queue.get()
        if req is None:
            break
        xt = [torch.tensor(x, dtype=torch.int64, device=device).reshape(-1, 1)
              for _, x in req]
        outs = gen_sample_batched(model, xt, k=k, maxlen=maxlen,
                                  use_unk=True, kl_factor=kl_factor,
                                  ctx_factor=ctx_factor,
                                  state_factor=state_factor)
        for (idx, _), (sample, score, alphas) in zip(req, outs):
            score = numpy.array(score)
            if normalize:
                lengths = numpy.array([len(s2) for s2 in sample])
                score = score / lengths
            sidx = int(numpy.argmin(score))
            rqueue.put((idx, sample[sidx],
                        [int(numpy.argmax(a)) for a in alphas[sidx]]))


def seqs2words(caps, pos, word_idict):
    """id sequences + positions -> "word [pos]" lines (gen.py:88-98)."""
    capsw = []
    for cc, pp in zip(caps, pos):
        ww = []
        for w, p in zip(cc, pp):
            if w == 0:
                break
            # ids in [len(dict), n_words) are valid model outputs when the
            # corpus vocabulary is smaller than n_words (a briefly-trained
            # model emits them); map to UNK like the trainer's sample
            # printer rather than crashing (the reference would KeyError
            # here, gen.py:93 — only because its corpora are larger than
            # its n_words caps)
            ww.append(word_idict.get(w, "UNK"))
            ww.append("[{0}]".format(p))
        capsw.append(" ".join(ww))
    return capsw


def map_line(line, word_dict, n_words, chr_level=False):
    """Tokenize + id-map one source line, appending eos (gen.py:100-109)."""
    if chr_level:
        words = list(line.strip())
    else:
        words = line.strip().split()
    x = [word_dict.get(w, 1) for w in words]
    x = [ii if ii < n_words else 1 for ii in x]
    x.append(0)
    return x


def generate_file(model_path, dictionary, source_file, saveto, k=5,
                  normalize=False, n_process=5, chr_level=False,
                  kl_factor=0.0, ctx_factor=0.0, state_factor=0.0,
                  devices=None, verbose=True, maxlen=100):
    """gen.py main() equivalent."""
    options = load_options(model_path)
    word_dict = load_dictionary(dictionary)
    word_idict = invert_dictionary(word_dict, with_specials=True)

    jobs = []
    with open(source_file) as f:
        for idx, line in enumerate(f):
            jobs.append((idx, map_line(line, word_dict, options["n_words"],
                                       chr_level)))
    n_samples = len(jobs)

    if devices is None:
        if torch.cuda.is_available():
            devices = ["cuda:%d" % i for i in range(torch.cuda.device_count())]
        else:
            devices = ["cpu"]

    trans = [None] * n_samples
    pos = [None] * n_samples

    if n_process <= 1:
        model = _make_model(model_path, options, devices[0])
        # batch several sentences' beams through each decode step (the
        # fused decoder kernels take up to 32 rows)
        sent_batch = max(1, 64 // max(k, 1))
        from .batched import gen_sample_batched
        for base in range(0, n_samples, sent_batch):
            chunk = jobs[base:base + sent_batch]
            xt = [torch.tensor(x, dtype=torch.int64,
                               device=devices[0]).reshape(-1, 1)
                  for _, x in chunk]
            outs = gen_sample_batched(model, xt, k=k, maxlen=maxlen,
                                      use_unk=True, kl_factor=kl_factor,
                                      ctx_factor=ctx_factor,
                                      state_factor=state_factor)
            for (idx, _), (sample, score, alphas) in zip(chunk, outs):
                score = numpy.array(score)
                if normalize:
                    lengths = numpy.array([len(s2) for s2 in sample])
                    score = score / lengths
                sidx = int(numpy.argmin(score))
                trans[idx] = sample[sidx]
                pos[idx] = [int(numpy.argmax(a)) for a in alphas[sidx]]
            if verbose and (base // sent_batch) % 4 == 0:
                print("Sample %d / %d Done" % (base + 1, n_samples))
    else:
        import torch.multiprocessing as mp
        ctx = mp.get_context("spawn")
        queue, rqueue = ctx.Queue(), ctx.Queue()
        procs = []
        for midx in range(n_process):
            dev = devices[midx % len(devices)]
            p = ctx.Process(target=translate_worker,
                            args=(queue, rqueue, midx, model_path, options, k,
                                  normalize, kl_factor, ctx_factor,
                                  state_factor, dev, maxlen))
            p.start()
            procs.append(p)
        sent_batch = max(1, 64 // max(k, 1))
        for base in range(0, n_samples, sent_batch):
            queue.put(jobs[base:base + sent_batch])
        for _ in range(n_process):
            queue.put(None)
        for i in range(n_samples):
            idx, seq, p_ = rqueue.get()
            trans[idx], pos[idx] = seq, p_
            if verbose and i % 10 == 0:
                print("Sample %d / %d Done" % (i + 1, n_samples))
        for p in procs:
            p.join()

    lines = seqs2words(trans, pos, word_idict)
    with open(saveto, "w") as f:
        f.write("\n".join(lines))
        f.write("\n")
    if verbose:
        print("Done")
