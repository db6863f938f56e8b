import numpy as np
rng = np.random.default_rng(5)
sents = []
for _ in range(3000):
    head = "aa" if rng.random() < 0.5 else "a2"
    filler = [f"x{rng.integers(0, 200)}" for _ in range(4)]
    sents.append([head, "bb"] + filler + [head, "bb"])
from glint_word2vec_amd import GlintWord2Vec

def run(device, atomic=True):
    est = (GlintWord2Vec().setVectorSize(32).setMinCount(1).setSeed(4)
           .setNumIterations(6).setWindowSize(2).setN(5)
           .setUnigramTableSize(100000).setStepSize(0.05)
           .setSubsampleRatio(0.0))
    est.config.device = device
    est.config.atomic_updates = atomic
    m = est.fit(sents)
    f = m.syn0 / np.linalg.norm(m.syn0, axis=1, keepdims=True)
    v = m.vocab
    sim_a2 = f[v["aa"]] @ f[v["a2"]]
    sims_x = [f[v["aa"]] @ f[v[f"x{i}"]] for i in range(200) if f"x{i}" in v]
    aa = v.index["aa"]
    norm_aa = np.linalg.norm(m.syn0[aa])
    s1n = np.linalg.norm(m.syn1)
    print(device, "atomic=", atomic, "sim(aa,a2)=%.3f" % sim_a2,
          "mean_x=%.3f" % np.mean(sims_x), "max_x=%.3f" % np.max(sims_x),
          "norm_aa=%.3f" % norm_aa, "syn1_norm=%.2f" % s1n)

run("cpu")
run("cuda", atomic=True)
run("cuda", atomic=False)
