# Developer targets (parity with the reference Makefile:5-28)
.PHONY: test test-gpu build bench mlflow tensorboard lint

test:
	python -m pytest tests/ -q -m "not gpu"

test-gpu:
	python -m pytest tests/ -q -m gpu

build:
	python -m torch_actor_critic_amd.ops.build

bench:
	python bench.py --steps 2000 --warmup 200

# MLflow UI over the local mlruns/ store (requires mlflow installed)
mlflow:
	mlflow ui --backend-store-uri ./mlruns

tensorboard:
	tensorboard --logdir runs/

lint:
	python -m flake8 torch_actor_critic_amd sac networks buffer environments tests main.py run_agent.py bench.py

# Reference-parity convenience targets (reference Makefile:17-28)
dvc:
	dvc repro

prefect:
	prefect server start
