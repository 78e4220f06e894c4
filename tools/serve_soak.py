"""Serving soak: fire N mixed requests at a running cake_amd.serve
instance and verify every response (status, token count, determinism of
seeded sampling).  Run on the GPU box next to `python -m cake_amd.serve`."""
import json
import sys
import urllib.request

BASE = sys.argv[1] if len(sys.argv) > 1 else "http://127.0.0.1:8000"
N = int(sys.argv[2]) if len(sys.argv) > 2 else 200


def post(body):
    req = urllib.request.Request(
        BASE + "/v1/chat/completions",
        data=json.dumps(body).encode(),
        headers={"Content-Type": "application/json"})
    with urllib.request.urlopen(req, timeout=120) as r:
        assert r.status == 200, r.status
        return json.loads(r.read())


def main():
    import random
    rnd = random.Random(7)
    seeded = {}
    fails = 0
    for i in range(N):
        mode = i % 4
        plen = rnd.randint(4, 96)
        ids = [rnd.randrange(0, 1000) for _ in range(plen)]
        mt = rnd.randint(4, 24)
        body = {"prompt_token_ids": ids, "max_tokens": mt}
        if mode == 1:
            body.update(temperature=0.8, seed=i)
        elif mode == 2:
            body.update(temperature=0.9, top_p=0.9, seed=1234)
        elif mode == 3:
            body.update(temperature=0.7, top_k=20, seed=i)
        out = post(body)
        toks = out["choices"][0]["token_ids"]
        if not (1 <= len(toks) <= mt):
            fails += 1
            print(f"[{i}] BAD token count {len(toks)} (max {mt})")
        if mode == 2:
            key = tuple(ids)
            if key in seeded and seeded[key] != tuple(toks):
                fails += 1
                print(f"[{i}] seeded top-p NOT reproducible")
            seeded[key] = tuple(toks)
        if (i + 1) % 50 == 0:
            print(f"{i + 1}/{N} ok so far, fails={fails}", flush=True)
    print(f"SOAK {'FAILED' if fails else 'PASSED'}: {N} requests, "
          f"{fails} failures")
    sys.exit(1 if fails else 0)


if __name__ == "__main__":
    main()
