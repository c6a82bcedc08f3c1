#!/bin/bash
# Canonical NCUP things fine-tune (reference train_raft_nc_things.sh hyperparameters), launched as
# one process per GPU over RCCL/xGMI. NGPU defaults to all visible GPUs.
set -e
mkdir -p checkpoints
NGPU=${NGPU:-$(python -c 'import torch; print(max(torch.cuda.device_count(), 1))')}

EXP=raft_nc533_things_ft

python -m torch.distributed.run --nnodes=1 --nproc-per-node $NGPU \
--master-addr 127.0.0.1 --master-port ${MASTER_PORT:-29531} train.py \
--name $EXP \
--model raft_nc_dbl \
--load_pretrained models/raft-things.pth \
--stage things \
--validation sintel \
--compressed_ft \
--num_steps 100000 \
--lr 0.000125 \
--image_size 400 720 \
--gpus 0 1 \
--batch_size 6 \
--optimizer adamW \
--scheduler cyclic \
--mixed_precision \
--final_upsampling=NConvUpsampler \
--final_upsampling_scale=4 \
--final_upsampling_use_data_for_guidance=True \
--final_upsampling_channels_to_batch=True \
--final_upsampling_use_residuals=False \
--final_upsampling_est_on_high_res=False \
--interp_net=NConvUNet \
--interp_net_channels_multiplier=2 \
--interp_net_num_downsampling=1 \
--interp_net_data_pooling="conf_based" \
--interp_net_encoder_filter_sz=5 \
--interp_net_decoder_filter_sz=3 \
--interp_net_out_filter_sz=1 \
--interp_net_shared_encoder=True \
--interp_net_use_double_conv=False \
--interp_net_use_bias=False \
--weights_est_net=Simple \
--weights_est_net_num_ch="[64, 32]" \
--weights_est_net_filter_sz="[3, 3, 1]" \
--weights_est_net_dilation="[1, 1, 1]"
