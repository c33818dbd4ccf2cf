"""End-to-end pipeline smoke tests on CPU (gloo)."""

import sys

import pytest
import torch

from dmlcloud_amd import TrainingPipeline, TrainValStage


class DummyDataset(torch.utils.data.Dataset):
    def __len__(self):
        return 8

    def __getitem__(self, idx):
        return torch.randn(10), torch.randint(0, 10, size=(1,)).item()


class DummyStage(TrainValStage):
    def pre_stage(self):
        self.model = torch.nn.Linear(10, 10)
        self.pipeline.register_model('linear', self.model)

        self.optimizer = torch.optim.SGD(self.model.parameters(), lr=1e-3)
        self.pipeline.register_optimizer('sgd', self.optimizer)

        self.pipeline.register_dataset('train', torch.utils.data.DataLoader(DummyDataset(), batch_size=4))
        self.pipeline.register_dataset('val', torch.utils.data.DataLoader(DummyDataset(), batch_size=4))

        self.loss = torch.nn.CrossEntropyLoss()

    def step(self, batch):
        x, y = batch
        x, y = x.to(self.device), y.to(self.device)
        model = self.pipeline.models['linear']
        output = model(x)
        return self.loss(output, y)


class TestSmoke:
    def test_smoke(self, torch_distributed):
        pipeline = TrainingPipeline()
        pipeline.append_stage(DummyStage(), max_epochs=1)
        pipeline.run()

        # standard metrics exist and were reduced
        tracker = pipeline.tracker
        assert tracker['train/loss'][0] is not None
        assert tracker['val/loss'][0] is not None
        assert tracker['misc/total_train_batches'][0].item() == 2
        assert tracker['misc/step_time_ms'][0] is not None

    def test_multi_epoch(self, torch_distributed):
        pipeline = TrainingPipeline()
        pipeline.append_stage(DummyStage(), max_epochs=3)
        pipeline.run()
        assert len(pipeline.tracker['train/loss']) == 3

    def test_two_stages_share_tracker(self, torch_distributed):
        """Multiple stages run in order on one tracker; the second stage
        reuses the registries the first one filled (warmup -> finetune)."""

        class FinetuneStage(DummyStage):
            def pre_stage(self):
                # model/optimizer/datasets already registered by stage 1
                self.loss = torch.nn.CrossEntropyLoss()

        pipeline = TrainingPipeline()
        pipeline.append_stage(DummyStage(), max_epochs=2, name='warmup')
        pipeline.append_stage(FinetuneStage(), max_epochs=1, name='finetune')
        pipeline.run()
        # 2 + 1 epochs of train/loss landed in one shared history
        assert len(pipeline.tracker['train/loss']) == 3
        assert all(v is not None for v in pipeline.tracker['train/loss'])
        assert pipeline.stages[0].stop_time is not None
        assert pipeline.stages[1].current_epoch == 2  # ran its single epoch

    def test_no_stages_raises(self, torch_distributed):
        pipeline = TrainingPipeline()
        with pytest.raises(ValueError):
            pipeline.run()

    def test_uninitialized_dist_raises(self):
        pipeline = TrainingPipeline()
        pipeline.append_stage(DummyStage(), max_epochs=1)
        with pytest.raises(ValueError):
            pipeline.run()

    def test_stop_stage(self, torch_distributed):
        class StopStage(DummyStage):
            def post_epoch(self):
                self.stop_stage()

        pipeline = TrainingPipeline()
        pipeline.append_stage(StopStage(), max_epochs=10)
        pipeline.run()
        assert len(pipeline.tracker['train/loss']) == 1

    def test_config_accessible(self, torch_distributed):
        pipeline = TrainingPipeline(config={'lr': 0.1, 'opt': {'name': 'sgd'}}, name='cfg-test')
        assert pipeline.config.lr == 0.1
        assert pipeline.config.opt.name == 'sgd'

    def test_gradient_clipping(self, torch_distributed):
        class ClipStage(DummyStage):
            def gradient_clip(self):
                return 0.5

        pipeline = TrainingPipeline()
        pipeline.append_stage(ClipStage(), max_epochs=1)
        pipeline.run()
        assert pipeline.tracker['train/loss'][0] is not None

    def test_flat_replica_path(self, torch_distributed):
        """The flat-buffer fast path trains end-to-end on CPU."""
        from dmlcloud_amd.parallel import FlatAdam

        class FlatStage(DummyStage):
            def pre_stage(self):
                model = torch.nn.Linear(10, 10)
                self.pipeline.register_model('linear', model, ddp_impl='flat')
                replica = self.pipeline.models['linear']
                self.pipeline.register_optimizer('adam', FlatAdam(replica, lr=1e-3))
                self.pipeline.register_dataset('train', torch.utils.data.DataLoader(DummyDataset(), batch_size=4))
                self.pipeline.register_dataset('val', torch.utils.data.DataLoader(DummyDataset(), batch_size=4))
                self.loss = torch.nn.CrossEntropyLoss()

        pipeline = TrainingPipeline()
        pipeline.append_stage(FlatStage(), max_epochs=2)
        pipeline.run()
        assert pipeline.tracker['train/loss'][1] is not None


if __name__ == '__main__':
    sys.exit(pytest.main([__file__]))
