# dcr_amd — common targets
.PHONY: build test test-gpu bench bench8 sanitize validate-drafts status clean

build:            ## compile the gfx950 HIP extension in-tree
	python -c "import __graft_entry__ as g; g.build()"

test:             ## CPU test suite (no GPU required)
	python -m pytest tests -q -m "not gpu"

test-gpu:         ## GPU suite (run on an MI355X box)
	python -m pytest tests -q -m gpu

bench:            ## flagship 1-GPU benchmark (SD-2.1 256px finetune)
	python bench.py --gpus 1 --steps 10 --warmup 6

bench8:           ## 8-GPU weak-scaling bench (one rank per GPU over RCCL)
	python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
	  --master-addr 127.0.0.1 bench.py --gpus 8 --steps 10 --warmup 6

sanitize:         ## kernel tests with serialized launches (GPU box)
	bash scripts/sanitize.sh

validate-drafts:  ## hardware-validate the gated round-2 draft kernels (GPU box)
	bash scripts/validate_drafts.sh

clean:
	rm -rf dcr_amd/ops/_build dcr_amd/ops/_dcr_hip.so

status:           ## build/kernel/env-gate status report
	python -m dcr_amd
