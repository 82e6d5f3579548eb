@echo off
REM Parity: reference scripts/start_server.bat (C23). ROCm serving is
REM Linux-only; this starts the CPU dev server on Windows.
python -m dts_amd.server --model llama-tiny --device cpu --host 127.0.0.1 --port 8000
