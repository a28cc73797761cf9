# End-to-end trainer check on one GPU: train with eval+save, then resume.
# Uses a small model and /tmp for checkpoints (they are large and scratch).
set -x
export LPP_WATCHDOG_S=600
timeout 600 python -m lpp_amd.trainer --config conf/llama_7b_pp4.yaml \
  model.name=llama-7b model.num_layers=4 num_stages=1 micro_batch_size=1 \
  gradient_accumulation_steps=4 seq_len=1024 \
  max_steps=6 save_steps=3 eval_steps=2 logging_steps=1 total_dataset_len=256 \
  output_dir=/tmp/tr_run 2>&1 | tail -8
timeout 300 python -m lpp_amd.trainer --config conf/llama_7b_pp4.yaml \
  model.name=llama-7b model.num_layers=4 num_stages=1 micro_batch_size=1 \
  gradient_accumulation_steps=4 seq_len=1024 \
  max_steps=8 save_steps=0 logging_steps=1 total_dataset_len=256 \
  output_dir=/tmp/tr_run2 resume=/tmp/tr_run/global_step6 2>&1 | tail -4
