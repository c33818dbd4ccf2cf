"""Pipeline checkpoint round-trip, graphs fallback, tracing, misc."""

import sys

import pytest
import torch

from dmlcloud_amd import TrainingPipeline, TrainValStage


class DS(torch.utils.data.Dataset):
    def __len__(self):
        return 8

    def __getitem__(self, idx):
        g = torch.Generator().manual_seed(idx)
        return torch.randn(10, generator=g), idx % 10


class Stage_(TrainValStage):
    def pre_stage(self):
        torch.manual_seed(0)
        model = torch.nn.Linear(10, 10)
        self.pipeline.register_model('m', model, save_interval=1)
        self.pipeline.register_optimizer('sgd', torch.optim.SGD(model.parameters(), lr=1e-2))
        self.pipeline.register_dataset('train', torch.utils.data.DataLoader(DS(), batch_size=4))
        self.pipeline.register_dataset('val', torch.utils.data.DataLoader(DS(), batch_size=4))
        self.loss = torch.nn.CrossEntropyLoss()

    def step(self, batch):
        x, y = batch
        return self.loss(self.pipeline.models['m'](x), y)


class TestCheckpointIntegration:
    def test_model_files_written(self, torch_distributed, tmp_path):
        pipeline = TrainingPipeline(name='ckpt')
        pipeline.enable_checkpointing(str(tmp_path), resume=False)
        pipeline.append_stage(Stage_(), max_epochs=2)
        pipeline.run()

        ckpt = pipeline.checkpoint_dir
        assert ckpt.is_valid
        assert ckpt.config_file.exists()
        model_dir = ckpt.models_dir / 'm'
        assert (model_dir / 'latest.dmlt').exists()
        assert (model_dir / 'epoch_0001.dmlt').exists()
        assert (model_dir / 'epoch_0002.dmlt').exists()

    def test_full_state_roundtrip(self, torch_distributed, tmp_path):
        pipeline = TrainingPipeline(name='ckpt')
        pipeline.enable_checkpointing(str(tmp_path), resume=False)
        stage = Stage_()
        pipeline.append_stage(stage, max_epochs=2)
        pipeline.run()
        pipeline.save_checkpoint()
        weight = pipeline.models['m'].module.weight.detach().clone()
        loss_history = pipeline.tracker['train/loss']

        # fresh pipeline resumes from the saved state
        pipeline2 = TrainingPipeline(name='ckpt2')
        pipeline2.checkpoint_dir = pipeline.checkpoint_dir
        pipeline2.device = torch.device('cpu')
        stage2 = Stage_()
        pipeline2.append_stage(stage2, max_epochs=2)
        stage2.pre_stage()
        pipeline2.load_checkpoint()
        torch.testing.assert_close(pipeline2.models['m'].module.weight.detach(), weight)
        assert len(pipeline2.tracker['train/loss']) == len(loss_history)
        assert stage2.current_epoch == stage.current_epoch

    def test_resume_flag(self, torch_distributed, tmp_path):
        p1 = TrainingPipeline(name='r')
        p1.enable_checkpointing(str(tmp_path), resume=False)
        p1.append_stage(Stage_(), max_epochs=1)
        p1.run()
        path = str(p1.checkpoint_dir.path)

        p2 = TrainingPipeline(name='r')
        p2.enable_checkpointing(path, resume=True)
        assert p2.resumed is True

        p3 = TrainingPipeline(name='r')
        p3.enable_checkpointing(str(tmp_path / 'other'), resume=True)
        assert p3.resumed is False


class TestFlatCheckpointRoundtrip:
    def test_bf16_flat_state_roundtrip(self, torch_distributed, tmp_path):
        """bf16 flat replica + FlatAdam full-state save/load restores the
        fp32 master exactly."""
        from dmlcloud_amd.checkpoint import load_tensor_state, save_tensor_state
        from dmlcloud_amd.parallel import FlatAdam, FlatReplica

        torch.manual_seed(0)
        model = torch.nn.Linear(10, 6)
        replica = FlatReplica(model, dtype=torch.bfloat16)
        opt = FlatAdam(replica, lr=1e-2)
        x = torch.randn(4, 10).to(torch.bfloat16)
        for _ in range(2):
            replica.zero_grad()
            replica(x).float().pow(2).mean().backward()
            opt.step()

        state = {'model': replica.state_dict(), 'opt': opt.state_dict()}
        path = tmp_path / 's.dmlt'
        save_tensor_state(state, path)
        loaded = load_tensor_state(path)

        model2 = torch.nn.Linear(10, 6)
        replica2 = FlatReplica(model2, dtype=torch.bfloat16)
        opt2 = FlatAdam(replica2, lr=1e-2)
        replica2.load_state_dict(loaded['model'])
        opt2.load_state_dict(loaded['opt'])

        torch.testing.assert_close(replica2.flat_master, replica.flat_master)
        torch.testing.assert_close(replica2.flat_param, replica.flat_param)
        torch.testing.assert_close(opt2.exp_avg, opt.exp_avg)
        assert opt2.step_t.item() == 2

    def test_pipeline_load_refreshes_flat_master(self, torch_distributed, tmp_path):
        """pipeline.load_checkpoint must route through FlatReplica's own
        load_state_dict so the fp32 master follows the loaded weights even
        when no optimizer state is restored for that model."""
        from dmlcloud_amd.parallel import FlatAdam

        class FlatStage(Stage_):
            def pre_stage(self):
                torch.manual_seed(self.pipeline.config.get('seed', 0))
                model = torch.nn.Linear(10, 10)
                self.pipeline.register_model('m', model, ddp_impl='flat', flat_dtype=torch.bfloat16)
                replica = self.pipeline.models['m']
                self.pipeline.register_optimizer('adam', FlatAdam(replica, lr=1e-3))
                self.pipeline.register_dataset('train', torch.utils.data.DataLoader(DS(), batch_size=4))
                self.pipeline.register_dataset('val', torch.utils.data.DataLoader(DS(), batch_size=4))
                self.loss = torch.nn.CrossEntropyLoss()

            def step(self, batch):
                x, y = batch
                return self.loss(self.pipeline.models['m'](x.to(torch.bfloat16)).float(), y)

        pipeline = TrainingPipeline(config={'seed': 0}, name='flatckpt')
        pipeline.enable_checkpointing(str(tmp_path), resume=False)
        pipeline.append_stage(FlatStage(), max_epochs=1)
        pipeline.run()
        pipeline.save_checkpoint()
        trained_param = pipeline.models['m'].flat_param.clone()

        # fresh pipeline with DIFFERENT init; drop the optimizer state so
        # only the model load can refresh the master
        pipeline2 = TrainingPipeline(config={'seed': 123}, name='flatckpt')
        pipeline2.enable_checkpointing(str(tmp_path / 'other'), resume=False)
        stage2 = FlatStage()
        pipeline2.append_stage(stage2, max_epochs=1)
        pipeline2.run()
        pipeline2.checkpoint_dir = pipeline.checkpoint_dir
        pipeline2.optimizers.clear()
        pipeline2.load_checkpoint()

        replica2 = pipeline2.models['m']
        torch.testing.assert_close(replica2.flat_param, trained_param)
        torch.testing.assert_close(replica2.flat_master, replica2.flat_param.to(torch.float32))


class TestGraphedStepCpu:
    def test_eager_fallback(self):
        from dmlcloud_amd.parallel import GraphedStep

        calls = []
        gs = GraphedStep(lambda: calls.append(1), enabled=False)
        gs.initialize()
        assert not gs.captured
        gs()
        gs()
        assert len(calls) == 2


class TestTracing:
    def test_roctx_noop_cpu(self):
        from dmlcloud_amd.utils.tracing import enable_tracing, roctx_range

        enable_tracing(True)
        try:
            with roctx_range('x'):
                pass
        finally:
            enable_tracing(False)
        with roctx_range('y'):
            pass


class TestFlatBf16Cpu:
    def test_bf16_pipeline_flat(self, torch_distributed):
        from dmlcloud_amd.parallel import FlatAdam

        class FlatStage(Stage_):
            def pre_stage(self):
                torch.manual_seed(0)
                model = torch.nn.Linear(10, 10)
                self.pipeline.register_model('m', model, ddp_impl='flat', flat_dtype=torch.bfloat16)
                replica = self.pipeline.models['m']
                self.pipeline.register_optimizer('adam', FlatAdam(replica, lr=1e-3))
                self.pipeline.register_dataset('train', torch.utils.data.DataLoader(DS(), batch_size=4))
                self.pipeline.register_dataset('val', torch.utils.data.DataLoader(DS(), batch_size=4))
                self.loss = torch.nn.CrossEntropyLoss()

            def step(self, batch):
                x, y = batch
                out = self.pipeline.models['m'](x.to(torch.bfloat16))
                return self.loss(out.float(), y)

        pipeline = TrainingPipeline()
        pipeline.append_stage(FlatStage(), max_epochs=1)
        pipeline.run()
        assert pipeline.tracker['train/loss'][0] is not None


class TestFlatLRScheduling:
    """FlatOptimizer is a real torch.optim.Optimizer: torch LR schedulers
    drive the fused step and misc/lr_* tracks."""

    def test_flat_optimizer_is_torch_optimizer(self, torch_distributed):
        from dmlcloud_amd.parallel import FlatAdam, FlatReplica, FlatSGD

        replica = FlatReplica(torch.nn.Linear(4, 2))
        assert isinstance(FlatSGD(replica, lr=0.1), torch.optim.Optimizer)
        replica2 = FlatReplica(torch.nn.Linear(4, 2))
        opt = FlatAdam(replica2, lr=1e-3)
        assert isinstance(opt, torch.optim.Optimizer)
        assert opt.param_groups[0]['lr'] == pytest.approx(1e-3)
        assert opt.lr == pytest.approx(1e-3)

    def test_steplr_drives_fused_step(self, torch_distributed):
        from dmlcloud_amd.parallel import FlatReplica, FlatSGD

        torch.manual_seed(0)
        replica = FlatReplica(torch.nn.Linear(4, 2))
        opt = FlatSGD(replica, lr=0.1)
        sched = torch.optim.lr_scheduler.StepLR(opt, step_size=1, gamma=0.5)

        lrs = []
        for _ in range(3):
            replica.zero_grad()
            replica(torch.ones(2, 4)).sum().backward()
            opt.step()
            lrs.append(opt.param_groups[0]['lr'])
            sched.step()
        assert lrs == pytest.approx([0.1, 0.05, 0.025])

    def test_pipeline_tracks_scheduled_lr(self, torch_distributed):
        from dmlcloud_amd.parallel import FlatSGD

        class SchedStage(Stage_):
            def pre_stage(self):
                torch.manual_seed(0)
                model = torch.nn.Linear(10, 10)
                self.pipeline.register_model('m', model, ddp_impl='flat')
                replica = self.pipeline.models['m']
                opt = FlatSGD(replica, lr=0.1)
                sched = torch.optim.lr_scheduler.StepLR(opt, step_size=1, gamma=0.1)
                self.pipeline.register_optimizer('opt', opt, sched)
                self.pipeline.register_dataset('train', torch.utils.data.DataLoader(DS(), batch_size=4))
                self.pipeline.register_dataset('val', torch.utils.data.DataLoader(DS(), batch_size=4))
                self.loss = torch.nn.CrossEntropyLoss()

        pipeline = TrainingPipeline()
        pipeline.append_stage(SchedStage(), max_epochs=3)
        pipeline.run()
        lr_history = pipeline.tracker['misc/lr_opt']
        assert lr_history[0] == pytest.approx(0.1)
        assert lr_history[1] == pytest.approx(0.01)
        assert lr_history[2] == pytest.approx(0.001)


class TestFusedCEIgnoreIndex:
    def test_cpu_fallback_matches_torch(self):
        from dmlcloud_amd.ops.fused_loss import cross_entropy

        torch.manual_seed(0)
        logits = torch.randn(6, 11)
        targets = torch.tensor([0, 3, -100, 5, -100, 10])
        expected = torch.nn.functional.cross_entropy(logits, targets)
        torch.testing.assert_close(cross_entropy(logits, targets), expected)


if __name__ == '__main__':
    sys.exit(pytest.main([__file__]))
