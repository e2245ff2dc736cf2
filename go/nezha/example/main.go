//go:build cgo

// Example: train the flagship MLP from Go (package nezha over the C ABI).
// Build where a Go toolchain exists (none in this image — SURVEY.md §0.2):
//
//	cd go/nezha/example && CGO_CFLAGS="-I$REPO/include" \
//	  CGO_LDFLAGS="-L$REPO/build -lnezha" go run .
package main

import (
	"fmt"
	"log"

	nezha "github.com/fast-ml/nezha-amd/go/nezha"
)

func main() {
	dev, idx := nezha.CPU, 0
	if nezha.HipDeviceCount() > 0 {
		dev = nezha.HIP
	}
	ctx, err := nezha.NewCtx(dev, idx)
	if err != nil {
		log.Fatal(err)
	}
	tr, err := nezha.NewMLPTrainer(ctx, 4, 4096, 8192, "adam", 1e-4, 64<<20, 1234)
	if err != nil {
		log.Fatal(err)
	}
	fmt.Printf("params: %d\n", tr.NumParams())
	for step := 1; step <= 100; step++ {
		if err := tr.Step(); err != nil {
			log.Fatal(err)
		}
		if step%20 == 0 {
			loss, _ := tr.Loss()
			fmt.Printf("step %3d  loss %.4f\n", step, loss)
		}
	}
	if err := tr.Save("mlp.npz"); err != nil {
		log.Fatal(err)
	}
	fmt.Println("saved mlp.npz")
}
