# dist_tuto_pth_amd — build + demo targets (the reference's Makefile:4-9
# exposed `all` (docs) and `ptp` (smoke run); this one adds the full
# build/test surface).

PY ?= python3

.PHONY: all build docs test test-gpu ptp sendrecv allreduce train bench clean

all: build docs

# render the tutorial to HTML (the reference's `all` target renders
# tuto.md -> tuto.html + index.html via its paperify.py)
docs:
	$(PY) docs/paperify.py

build:
	$(PY) build.py

test:
	$(PY) -m pytest tests/ -x -q -m "not gpu"

test-gpu:
	$(PY) -m pytest tests/ -x -q -m gpu

# the reference's `make ptp` smoke target (Makefile:8-9)
ptp:
	$(PY) examples/ptp.py

sendrecv:
	$(PY) examples/send_recv.py

allreduce:
	$(PY) examples/allreduce.py --world 4 --algo chunked

train:
	$(PY) examples/train_dist.py --world 2 --epochs 2

bench:
	$(PY) bench.py --steps 50 --warmup 10

clean:
	rm -rf dist_tuto_pth_amd/_native/*.so build __pycache__
