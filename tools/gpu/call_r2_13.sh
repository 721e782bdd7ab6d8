cd /root/repo
python -m pytest tests/test_gpu_decode.py::test_rich_scalar_store_gpu_route -q 2>&1 | tail -30
