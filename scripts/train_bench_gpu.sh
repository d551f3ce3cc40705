python -m petals_amd.cli.run_dht --host 127.0.0.1 --port 31901 > gpurun_out/dht_t.log 2>&1 &
DHT_PID=$!
sleep 2
python -m petals_amd.cli.run_server llama-2-7b --host 127.0.0.1 --initial_peers 127.0.0.1:31901 \
  --torch_dtype bfloat16 --dht_prefix train-bench --throughput 1000 > gpurun_out/srv_t.log 2>&1 &
SRV_PID=$!
for i in $(seq 1 60); do grep -q "listening on" gpurun_out/srv_t.log && break; sleep 2; done
timeout 300 python benchmarks/benchmark_training.py --model llama-2-7b --initial_peers 127.0.0.1:31901 \
  --dht_prefix train-bench --pre_seq_len 8 --tuning_mode deep_ptune --batch_size 2 --seq_len 64 --n_steps 6
RC=$?
kill $SRV_PID $DHT_PID 2>/dev/null
exit $RC
