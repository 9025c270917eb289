/*
Copyright 2021.

Licensed under the Apache License, Version 2.0 (the "License");
you may not use this file except in compliance with the License.
You may obtain a copy of the License at

    http://www.apache.org/licenses/LICENSE-2.0

Unless required by applicable law or agreed to in writing, software
distributed under the License is distributed on an "AS IS" BASIS,
WITHOUT WARRANTIES OR CONDITIONS OF ANY KIND, either express or implied.
See the License for the specific language governing permissions and
limitations under the License.
*/

package edgeplatform

import (
	"fmt"
	"os"

	"github.com/spf13/cobra"

	"github.com/acme/edgeplatform/apis/edgeplatform"

	v1alpha1edgeplatform "github.com/acme/edgeplatform/apis/edgeplatform/v1alpha1/edgeplatform"
	cmdinit "github.com/acme/edgeplatform/cmd/edge-platform-ctl/commands/init"
	//+operator-builder:imports
)

// getEdgePlatformManifest returns the sample EdgePlatform manifest
// based upon API Version input.
func getEdgePlatformManifest(i *cmdinit.InitSubCommand) (string, error) {
	apiVersion := i.APIVersion
	if apiVersion == "" || apiVersion == "latest" {
		return edgeplatform.EdgePlatformLatestSample, nil
	}

	// generate a map of all versions to samples for each api version created
	manifestMap := map[string]string{
		"v1alpha1": v1alpha1edgeplatform.Sample(i.RequiredOnly),
		//+operator-builder:versionmap
	}

	// return the manifest if it is not blank
	manifest := manifestMap[apiVersion]
	if manifest != "" {
		return manifest, nil
	}

	// return an error if we did not find a manifest for an api version
	return "", fmt.Errorf("unsupported API Version: " + apiVersion)
}

// NewEdgePlatformSubCommand creates a new command and adds it to its
// parent command.
func NewEdgePlatformSubCommand(parentCommand *cobra.Command) {
	initCmd := &cmdinit.InitSubCommand{
		Name:         "collection",
		Description:  "Manage edgeplatform workload",
		InitFunc:     InitEdgePlatform,
		SubCommandOf: parentCommand,
	}

	initCmd.Setup()
}

func InitEdgePlatform(i *cmdinit.InitSubCommand) error {
	manifest, err := getEdgePlatformManifest(i)
	if err != nil {
		return fmt.Errorf("unable to get manifest for EdgePlatform; %w", err)
	}

	outputStream := os.Stdout

	if _, err := outputStream.WriteString(manifest); err != nil {
		return fmt.Errorf("failed to write to stdout, %w", err)
	}

	return nil
}
