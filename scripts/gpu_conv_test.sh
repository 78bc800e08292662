export SCALERL_EXPERIMENTAL=1
timeout 150 python -m pytest tests/test_conv_experimental.py -m gpu -q 2>&1 | tail -6
