import sys, tempfile, torch
sys.path.insert(0, "/root/repo")
from mi355x_scale.data.generator import write_image_parquet
from mi355x_scale.train import ImageClassifier, ImageStreamDataModule, Trainer
import os
which = sys.argv[1] if len(sys.argv) > 1 else "all"
def graphtests():
    import copy
    from mi355x_scale.train.graphstep import GraphedTrainStep
    torch.manual_seed(0)
    dev = torch.device("cuda:0")
    model = ImageClassifier("resnet18", num_classes=10, lr=1e-2).to(dev)
    opt = torch.optim.Adam(model.parameters(), lr=1e-2, foreach=True, capturable=True)
    b = {"image": torch.randint(0, 256, (8, 64, 64, 3), dtype=torch.uint8, device=dev),
         "label": torch.randint(0, 10, (8,), device=dev)}
    gs = GraphedTrainStep(model, opt, b, world_size=1, warmup=2)
    for _ in range(5):
        gs.step(b)
    torch.cuda.synchronize()
    print("graphtest ok")
def trainertest():
    d = tempfile.mkdtemp()
    write_image_parquet(d, num_rows=64, image_hw=(64, 64), rows_per_group=16, rows_per_file=32)
    model = ImageClassifier("resnet18", num_classes=10, lr=1e-3)
    dm = ImageStreamDataModule(d, batch_size=16, workers_count=2, image_hw=(64, 64))
    trainer = Trainer(max_epochs=1, limit_train_batches=4, limit_val_batches=2,
                      default_root_dir=d + "/ckpt")
    trainer.fit(model, dm)
    print("trainertest ok")
if which in ("all", "graph"):
    graphtests()
if which in ("all", "trainer"):
    trainertest()
print("DONE", which)
