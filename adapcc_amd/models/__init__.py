"""Workload model zoo (GPT-2, ViT, VGG, ResNet, expert-parallel MoE)."""
