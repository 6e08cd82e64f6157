#!/usr/bin/env python3
"""A reference-style ASGD driver written against the ASYNC verb layer —
the code shape of the reference's SparkASGDThread main loop
running on this framework's AsyncRDD/ASYNCcontext/ASYNCbroadcast.

    python examples/verb_layer_driver.py
"""

import os
import sys
import threading
import time

import numpy as np

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

from asyncframework_amd.core import ASYNCcontext, AsyncRDD, ASYNCbroadcast

P, D, N = 4, 16, 2000
GAMMA, RATE, TAU, NUM_ITER = 0.05, 0.2, 1 << 30, 300

rng = np.random.default_rng(0)
X = rng.standard_normal((N, D))
y = X @ rng.standard_normal(D)
points = [(X[i], y[i]) for i in range(N)]
rdd = AsyncRDD([points[i * (N // P):(i + 1) * (N // P)] for i in range(P)])
AC = ASYNCcontext()

w = np.zeros(D)
k = 0
stop = threading.Event()


def gradfun(point, wv):
    x, yy = point
    return (x @ wv - yy) * x


def updater():  # the parameter-server thread (reference :153-226)
    global w, k
    while not stop.is_set():
        try:
            pr = AC.ASYNCcollectAll(timeout=0.2)
        except Exception:
            continue
        if pr.gettaskResult() is None:
            continue
        if pr.getStaleness() <= TAU:  # the tau filter (reference :172)
            w = w - GAMMA * pr.gettaskResult() / (RATE * N / P)
            k += 1


threading.Thread(target=updater, daemon=True).start()
t0 = time.perf_counter()
rounds = 0
while k < NUM_ITER and rounds < 5000:
    if AC.STAT:  # quorum gate (reference :233-237)
        if next(iter(AC.STAT.values())).getAvailableWorkers() < P // 2:
            time.sleep(0.0005)
            continue
    bc = ASYNCbroadcast(w.copy())  # versioned weights (reference :245)
    (rdd.ASYNCbarrier(lambda st: st.getAvailability(), AC.STAT)
        .sample(False, RATE, 42 + rounds + 1)
        .map(lambda p, _bc=bc: gradfun(p, _bc.value()))
        .ASYNCreduce(lambda a, b: a + b, AC))
    rounds += 1
stop.set()

obj = float(((X @ w - y) ** 2).mean())
print(f"k={k} rounds={rounds} objective={obj:.5f} "
      f"elapsed={time.perf_counter() - t0:.2f}s")
assert obj < 1.0
